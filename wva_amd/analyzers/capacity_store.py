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
