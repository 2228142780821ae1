"""Capacity knowledge store.

Parity: reference internal/engines/analyzers/saturation_v2/capacity_store.go
:16-187 — thread-safe map keyed "namespace|modelID|variantName" with live >
deployment precedence, staleness/eviction, and cross-namespace
FindCompatible matching on (model, accelerator, gpuCount, compatible vLLM
params), preferring live records.

MI355X note: records are keyed by gpuCount so TP=1/2/4/8 xGMI variants each
carry their own measured capacity (service rates change with TP
because decode includes per-layer RCCL all-reduce time over xGMI).
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Dict, Optional

from ..kube.objects import Deployment
from .deployment_parser import VLLMEngineParams, parse_vllm_args

# (constants.go:5-28)
ROLLING_AVERAGE_WINDOW_SIZE = 10
CAPACITY_STALENESS_TIMEOUT_S = 30 * 60.0
CAPACITY_EVICTION_TIMEOUT_S = 7 * 24 * 3600.0
HISTORY_EVICTION_TIMEOUT_S = 24 * 3600.0
BYTES_PER_TOKEN = 4


@dataclass
class CapacityRecord:
    accelerator_name: str = ""
    gpu_count: int = 1
    num_gpu_blocks: int = 0
    block_size: int = 0
    total_kv_capacity_tokens: int = 0
    effective_capacity: int = 0
    vllm_params: Optional[VLLMEngineParams] = None
    learned_from: str = ""  # "live" | "deployment" | "annotation"
    learned_at: float = field(default_factory=time.monotonic)


def _store_key(namespace: str, model_id: str, variant_name: str) -> str:
    return f"{namespace}|{model_id}|{variant_name}"


def normalize_kv_dtype(kv_cache_dtype: str) -> str:
    """vLLM --kv-cache-dtype → capacity family: fp8/fp8_e4m3/fp8_e5m2
    halve the bytes-per-token (double k1); auto/bf16/fp16 are the
    2-byte family."""
    return "fp8" if (kv_cache_dtype or "").startswith("fp8") else "bf16"


@dataclass
class MeasuredProfile:
    """A hardware-measured capacity point (profiles/calibration_*.json):
    keyed by (model, accelerator, gpuCount, kv dtype family). This is
    the MI355X-native improvement over the reference's
    deployment-derived guesswork — before any live `cache_config_info`
    arrives, a new variant's KV capacity comes from a real measurement
    on the actual device (288 GB HBM3E), fp8 giving its true 2× k1
    (measured 4.0M vs 2.0M tokens for Llama-3.1-8B)."""

    model_id: str
    accelerator_name: str
    gpu_count: int
    kv_dtype: str  # "bf16" | "fp8"
    num_gpu_blocks: int
    block_size: int
    total_kv_capacity_tokens: int


class CapacityKnowledgeStore:
    def __init__(self) -> None:
        self._lock = threading.RLock()
        self._records: Dict[str, CapacityRecord] = {}
        # (model, accelerator, gpu_count, kv_dtype) → MeasuredProfile
        self._measured: Dict[tuple, MeasuredProfile] = {}

    # --- measured profiles (MI355X calibration registry) ---

    def register_measured_profile(self, profile: MeasuredProfile) -> None:
        with self._lock:
            key = (
                profile.model_id,
                profile.accelerator_name,
                profile.gpu_count,
                profile.kv_dtype,
            )
            self._measured[key] = profile

    def measured_profile(
        self, model_id: str, accelerator: str, gpu_count: int, kv_dtype: str
    ) -> Optional[MeasuredProfile]:
        with self._lock:
            return self._measured.get(
                (model_id, accelerator, gpu_count, normalize_kv_dtype(kv_dtype))
            )

    def load_measured_profiles(
        self, records: list, accelerator: str = "MI355X"
    ) -> int:
        """Ingest calibration JSON dicts (profiles/calibration_*.json
        schema: model, gpu_count, num_gpu_blocks, block_size,
        kv_capacity_tokens, optional kv_dtype). Returns count loaded."""
        n = 0
        for rec in records:
            try:
                self.register_measured_profile(MeasuredProfile(
                    model_id=rec["model"],
                    accelerator_name=rec.get("accelerator", accelerator),
                    gpu_count=int(rec.get("gpu_count", 1)),
                    kv_dtype=normalize_kv_dtype(rec.get("kv_dtype", "bf16")),
                    num_gpu_blocks=int(rec["num_gpu_blocks"]),
                    block_size=int(rec.get("block_size", 16)),
                    total_kv_capacity_tokens=int(rec["kv_capacity_tokens"]),
                ))
                n += 1
            except (KeyError, TypeError, ValueError):
                continue
        return n

    def update(
        self, namespace: str, model_id: str, variant_name: str, record: CapacityRecord
    ) -> None:
        with self._lock:
            record.learned_at = time.monotonic()
            self._records[_store_key(namespace, model_id, variant_name)] = record

    def get(
        self, namespace: str, model_id: str, variant_name: str
    ) -> Optional[CapacityRecord]:
        with self._lock:
            return self._records.get(_store_key(namespace, model_id, variant_name))

    def is_stale(self, namespace: str, model_id: str, variant_name: str) -> bool:
        with self._lock:
            rec = self._records.get(_store_key(namespace, model_id, variant_name))
            if rec is None:
                return True
            return time.monotonic() - rec.learned_at > CAPACITY_STALENESS_TIMEOUT_S

    def load_from_deployment(
        self,
        namespace: str,
        model_id: str,
        variant_name: str,
        accelerator: str,
        gpu_count: int,
        deploy: Optional[Deployment],
    ) -> None:
        """Store a deployment-derived estimate; never overwrites live data."""
        if deploy is None:
            return
        with self._lock:
            key = _store_key(namespace, model_id, variant_name)
            existing = self._records.get(key)
            if existing is not None and existing.learned_from == "live":
                return
            params = parse_vllm_args(deploy)
            record = CapacityRecord(
                accelerator_name=accelerator,
                gpu_count=gpu_count,
                vllm_params=params,
                learned_from="deployment",
            )
            if params.num_gpu_blocks_override > 0:
                record.num_gpu_blocks = params.num_gpu_blocks_override
                record.block_size = params.block_size
                record.total_kv_capacity_tokens = (
                    params.num_gpu_blocks_override * params.block_size
                )
            else:
                # measured-profile fallback: a calibration record for
                # (model, accel, gpuCount, kv dtype) measured on real
                # hardware beats guessing — and carries the fp8 2× k1
                # (kv-cache-dtype=fp8 parsed by the deployment parser,
                # deployment_parser.go:182-219)
                measured = self._measured.get((
                    model_id, accelerator, gpu_count,
                    normalize_kv_dtype(params.kv_cache_dtype),
                ))
                if measured is not None:
                    record.num_gpu_blocks = measured.num_gpu_blocks
                    record.block_size = measured.block_size
                    record.total_kv_capacity_tokens = (
                        measured.total_kv_capacity_tokens
                    )
                    record.learned_from = "measured-profile"
            # Conservative floor so brand-new variants are still scale-up
            # candidates: the per-step token budget is a safe lower bound.
            if record.effective_capacity <= 0 and params.effective_max_batched_tokens > 0:
                record.effective_capacity = params.effective_max_batched_tokens
            self._records[key] = record

    def evict_stale(self, timeout_seconds: float) -> int:
        with self._lock:
            now = time.monotonic()
            stale = [
                k
                for k, r in self._records.items()
                if now - r.learned_at > timeout_seconds
            ]
            for k in stale:
                del self._records[k]
            return len(stale)

    def find_compatible(
        self,
        model_id: str,
        accelerator: str,
        gpu_count: int,
        params: Optional[VLLMEngineParams],
    ) -> Optional[CapacityRecord]:
        """Cross-namespace search for a capacity-equivalent record,
        preferring live over deployment-derived."""
        with self._lock:
            best: Optional[CapacityRecord] = None
            for key, rec in self._records.items():
                parts = key.split("|", 2)
                if len(parts) < 3 or parts[1] != model_id:
                    continue
                if rec.accelerator_name != accelerator or rec.gpu_count != gpu_count:
                    continue
                if rec.vllm_params is None or not rec.vllm_params.is_capacity_compatible(
                    params
                ):
                    continue
                if rec.effective_capacity <= 0 and rec.total_kv_capacity_tokens <= 0:
                    continue
                if best is None or (
                    best.learned_from != "live" and rec.learned_from == "live"
                ):
                    best = rec
            return best

    def __len__(self) -> int:
        with self._lock:
            return len(self._records)


def load_calibration_dir(
    store: CapacityKnowledgeStore, dirpath: str, accelerator: str = "MI355X"
) -> int:
    """Ingest every profiles/calibration_*.json under `dirpath` into the
    store's measured-profile registry. kv dtype is taken from an explicit
    `kv_dtype` field or inferred from an `_fp8` filename suffix (the
    calibration harness' naming). Returns the number of profiles loaded.
    """
    import glob
    import json
    import os

    records = []
    for path in sorted(glob.glob(os.path.join(dirpath, "calibration_*.json"))):
        try:
            with open(path) as f:
                d = json.load(f)
        except (OSError, ValueError):
            continue
        if "kv_capacity_tokens" not in d or "model" not in d:
            continue  # e.g. calibration_tp sweep summaries
        if "kv_dtype" not in d:
            d["kv_dtype"] = (
                "fp8" if "_fp8" in os.path.basename(path) else "bf16"
            )
        records.append(d)
    return store.load_measured_profiles(records, accelerator=accelerator)


# --- ConfigMap persistence (improvement over the reference) -----------------
#
# The reference's CapacityKnowledgeStore is memory-only: on controller
# restart every live-learned record is lost and zero-replica capacity
# estimates degrade until re-learned (SURVEY §5 checkpoint/resume:
# "In-memory-only state that is lost on restart ... optionally persist
# capacity records in a ConfigMap as an improvement"). These helpers
# serialize the store to the `wva-capacity-store` ConfigMap so a
# restarted controller resumes with its learned capacities, ages intact
# (eviction and staleness keep working across restarts).

CAPACITY_STORE_CONFIG_MAP_NAME = "wva-capacity-store"


def _params_to_dict(p: Optional[VLLMEngineParams]) -> Optional[dict]:
    if p is None:
        return None
    import dataclasses

    return dataclasses.asdict(p)


def _params_from_dict(d: Optional[dict]) -> Optional[VLLMEngineParams]:
    if not d:
        return None
    import dataclasses

    fields = {f.name for f in dataclasses.fields(VLLMEngineParams)}
    return VLLMEngineParams(**{k: v for k, v in d.items() if k in fields})


def snapshot_store(store: CapacityKnowledgeStore) -> dict:
    """Serializable snapshot: records with ages (learned_at is a
    monotonic clock, meaningless across processes)."""
    now = time.monotonic()
    with store._lock:
        return {
            "records": {
                key: {
                    "accelerator_name": r.accelerator_name,
                    "gpu_count": r.gpu_count,
                    "num_gpu_blocks": r.num_gpu_blocks,
                    "block_size": r.block_size,
                    "total_kv_capacity_tokens": r.total_kv_capacity_tokens,
                    "effective_capacity": r.effective_capacity,
                    "learned_from": r.learned_from,
                    "age_seconds": max(now - r.learned_at, 0.0),
                    "vllm_params": _params_to_dict(r.vllm_params),
                }
                for key, r in store._records.items()
            },
        }


def restore_store(store: CapacityKnowledgeStore, snap: dict) -> int:
    """Restore records from a snapshot; ages are preserved so staleness
    and eviction behave as if the process had never restarted. Never
    overwrites records learned in THIS process. Returns count restored."""
    now = time.monotonic()
    n = 0
    records = (snap or {}).get("records") or {}
    with store._lock:
        for key, d in records.items():
            if key in store._records:
                continue
            try:
                rec = CapacityRecord(
                    accelerator_name=str(d.get("accelerator_name", "")),
                    gpu_count=int(d.get("gpu_count", 1)),
                    num_gpu_blocks=int(d.get("num_gpu_blocks", 0)),
                    block_size=int(d.get("block_size", 0)),
                    total_kv_capacity_tokens=int(
                        d.get("total_kv_capacity_tokens", 0)
                    ),
                    effective_capacity=int(d.get("effective_capacity", 0)),
                    learned_from=str(d.get("learned_from", "")),
                    vllm_params=_params_from_dict(d.get("vllm_params")),
                )
            except (TypeError, ValueError):
                continue
            rec.learned_at = now - float(d.get("age_seconds", 0.0))
            store._records[key] = rec
            n += 1
    return n


class CapacityStorePersistence:
    """Periodic ConfigMap writer + bootstrap reader for the store."""

    def __init__(self, cluster, store: CapacityKnowledgeStore,
                 namespace: str, write_interval_seconds: float = 60.0,
                 analyzer=None):
        self.cluster = cluster
        self.store = store
        self.namespace = namespace
        self.write_interval_seconds = write_interval_seconds
        self.analyzer = analyzer  # optional: k2 history rides along
        self._last_write = 0.0
        self._last_payload = ""

    def restore(self) -> int:
        """Read the ConfigMap (if present) into the store."""
        import json

        cm = self.cluster.try_get(
            "ConfigMap", self.namespace, CAPACITY_STORE_CONFIG_MAP_NAME
        )
        if cm is None:
            return 0
        try:
            snap = json.loads(cm.data.get("records", "{}"))
        except ValueError:
            return 0
        n = restore_store(self.store, {"records": snap})
        if self.analyzer is not None and "history" in cm.data:
            try:
                hist = json.loads(cm.data["history"])
            except ValueError:
                hist = {}
            n += self.analyzer.history_restore(hist)
        return n

    def maybe_persist(self) -> bool:
        """Write the snapshot if the interval elapsed and something
        changed; best-effort (persistence failures never fail a tick)."""
        import json

        now = time.monotonic()
        if now - self._last_write < self.write_interval_seconds:
            return False
        snap = snapshot_store(self.store)
        payload = json.dumps(snap["records"], sort_keys=True)
        # change detection must ignore ages (they advance every call)
        canonical = json.dumps({
            k: {f: v for f, v in rec.items() if f != "age_seconds"}
            for k, rec in snap["records"].items()
        }, sort_keys=True)
        if canonical == self._last_payload:
            self._last_write = now
            return False
        from ..api.types import ObjectMeta
        from ..kube.objects import ConfigMap

        data = {"records": payload}
        if self.analyzer is not None:
            data["history"] = json.dumps(
                self.analyzer.history_snapshot(), sort_keys=True
            )
        cm = self.cluster.try_get(
            "ConfigMap", self.namespace, CAPACITY_STORE_CONFIG_MAP_NAME
        )
        try:
            if cm is None:
                self.cluster.create(ConfigMap(
                    metadata=ObjectMeta(
                        name=CAPACITY_STORE_CONFIG_MAP_NAME,
                        namespace=self.namespace,
                    ),
                    data=data,
                ))
            else:
                cm.data = data
                self.cluster.update(cm)
        except Exception:  # noqa: BLE001 — best-effort (incl. 409s)
            return False
        self._last_write = now
        self._last_payload = canonical
        return True
