"""V2 token-based saturation analyzer.

Parity: reference internal/engines/analyzers/saturation_v2/analyzer.go
:59-502. Capacity model:

  per replica:
    demand = tokensInUse + queueLen × avgInputTokens
    k1 = totalKvCapacityTokens × kvCacheThreshold        (memory-bound)
    k2 priority chain: observed-at-queue-saturation → rolling history
       (key model|accel|outputBucket) → derived from vLLM args
       (N_steady = min(B·O/(I+O), maxNumSeqs); k2 = N_steady·(I+O/2))
       → fallback k1                                     (compute-bound)
    effective = min(k1, k2)
  per variant: median effective over ready pods; zero-replica variants
    estimated from the capacity store or a compatible sibling
  model level:
    utilization = demand / supply
    requiredCapacity = demand/scaleUpThreshold − anticipatedSupply
    spareCapacity    = supply − demand/scaleDownBoundary
    + scheduler-queue demand:
      max(bytes/4, count×avgIn)×(1−prefixHitRate) + count×avgOut

MI355X note: k1 at 288 GB HBM3E reaches ~2.3× MI300X block counts; Python
ints are arbitrary-precision so no overflow guard is required (tested at
MI355X-scale block counts in tests/test_saturation_v2.py).
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional

from ..api.types import utcnow
from ..config.saturation import SaturationScalingConfig
from .capacity_store import (
    BYTES_PER_TOKEN,
    ROLLING_AVERAGE_WINDOW_SIZE,
    CapacityKnowledgeStore,
    CapacityRecord,
)
from .deployment_parser import VLLMEngineParams, classify_output_length
from .history import RollingAverage
from .interfaces import (
    AnalyzerInput,
    AnalyzerResult,
    ReplicaMetrics,
    SchedulerQueueMetrics,
    VariantCapacity,
    VariantReplicaState,
)

import dataclasses
import time


@dataclasses.dataclass
class ReplicaCapacity:
    pod_name: str = ""
    variant_name: str = ""
    accelerator_name: str = ""
    tokens_in_use: int = 0
    total_kv_capacity_tokens: int = 0
    memory_bound_capacity: int = 0
    compute_bound_capacity: int = 0
    effective_capacity: int = 0
    is_saturated: bool = False
    replica_demand: int = 0


def compute_model_workload_averages(replica_metrics: List[ReplicaMetrics]):
    """Model-level avg input/output tokens + prefix hit rate over replicas
    that report token stats."""
    avg_input = avg_output = avg_hit = 0.0
    count = 0
    for rm in replica_metrics:
        if rm.avg_input_tokens > 0 or rm.avg_output_tokens > 0:
            avg_input += rm.avg_input_tokens
            avg_output += rm.avg_output_tokens
            avg_hit += rm.prefix_cache_hit_rate
            count += 1
    if count:
        avg_input /= count
        avg_output /= count
        avg_hit /= count
    return avg_input, avg_output, avg_hit


def estimate_capacity_from_params(
    params: Optional[VLLMEngineParams], avg_input: float, avg_output: float
) -> int:
    """k2 derivation: N_steady = min(B·O/(I+O), S); k2 = N_steady·(I+O/2)."""
    if params is None or params.effective_max_batched_tokens <= 0 or avg_output <= 0:
        return 0
    B = float(params.effective_max_batched_tokens)
    S = float(params.max_num_seqs)
    I = avg_input
    O = avg_output
    n_steady = min(B * O / (I + O), S)
    k2 = int(n_steady * (I + O / 2))
    return k2 if k2 > 0 else 0


def estimate_scheduler_queue_demand(
    sq: Optional[SchedulerQueueMetrics],
    replica_metrics: List[ReplicaMetrics],
    drain_factor: float = 1.0,
) -> float:
    if sq is None or (sq.queue_size == 0 and sq.queue_bytes == 0):
        return 0.0
    avg_input, avg_output, avg_hit = compute_model_workload_averages(replica_metrics)
    input_tokens = max(sq.queue_bytes / BYTES_PER_TOKEN, sq.queue_size * avg_input)
    input_tokens *= 1 - avg_hit
    output_tokens = sq.queue_size * avg_output
    # drain_factor < 1 models that queued requests occupy capacity only for
    # their service time within the optimization interval (Little's law),
    # not as a standing concurrent population — see SaturationScalingConfig.
    return (input_tokens + output_tokens) * drain_factor


def _median(values: List[int]) -> int:
    n = len(values)
    if n == 0:
        return 0
    s = sorted(values)
    if n % 2 == 0:
        return (s[n // 2 - 1] + s[n // 2]) // 2
    return s[n // 2]


class SaturationAnalyzerV2:
    """Token-based analyzer selected by analyzerName == "saturation"."""

    def __init__(self, capacity_store: Optional[CapacityKnowledgeStore] = None):
        self._lock = threading.Lock()
        self._compute_capacity_history: Dict[str, RollingAverage] = {}
        # NOTE: explicit None check — CapacityKnowledgeStore defines __len__,
        # so an empty store is falsy and `or` would silently discard it.
        self.capacity_store = (
            capacity_store if capacity_store is not None else CapacityKnowledgeStore()
        )

    def name(self) -> str:
        return "saturation-token-based"

    def history_snapshot(self) -> dict:
        """Serializable k2 rolling-history state (checkpoint/resume
        companion to the capacity store: SURVEY §5 lists both as
        restart-lost)."""
        import time as _time

        now = _time.monotonic()
        with self._lock:
            return {
                key: {
                    "values": list(ra._values),
                    "age_seconds": max(now - ra.last_updated, 0.0),
                }
                for key, ra in self._compute_capacity_history.items()
            }

    def history_restore(self, snap: dict) -> int:
        """Restore k2 history; never overwrites locally observed keys;
        ages preserved for the 24 h eviction."""
        import time as _time

        from .history import RollingAverage
        from .capacity_store import ROLLING_AVERAGE_WINDOW_SIZE

        now = _time.monotonic()
        n = 0
        with self._lock:
            for key, d in (snap or {}).items():
                if key in self._compute_capacity_history:
                    continue
                try:
                    ra = RollingAverage(ROLLING_AVERAGE_WINDOW_SIZE)
                    for v in d.get("values", []):
                        ra._values.append(float(v))
                    ra.last_updated = now - float(d.get("age_seconds", 0.0))
                except (TypeError, ValueError):
                    continue
                self._compute_capacity_history[key] = ra
                n += 1
        return n

    def evict_stale_history(self, timeout_seconds: float) -> int:
        with self._lock:
            now = time.monotonic()
            stale = [
                k
                for k, ra in self._compute_capacity_history.items()
                if now - ra.last_updated > timeout_seconds
            ]
            for k in stale:
                del self._compute_capacity_history[k]
            return len(stale)

    # --- main entry ---

    def analyze(self, input: AnalyzerInput) -> AnalyzerResult:
        cfg = input.config
        if not isinstance(cfg, SaturationScalingConfig):
            raise TypeError(f"expected SaturationScalingConfig, got {type(cfg)}")

        gpus_by_variant = {
            vs.variant_name: vs.gpus_per_replica for vs in input.variant_states
        }

        # Phase 1: per-replica capacity
        replica_capacities: List[ReplicaCapacity] = []
        for rm in input.replica_metrics:
            rc = self._compute_replica_capacity(
                rm, cfg, input.model_id, input.namespace,
                gpus_by_variant.get(rm.variant_name, 1),
            )
            if rc is not None:
                replica_capacities.append(rc)

        # Phase 2: per-variant aggregation
        variant_capacities = self._aggregate_by_variant(
            replica_capacities,
            input.replica_metrics,
            input.variant_states,
            input.model_id,
            input.namespace,
            cfg.kv_cache_threshold,
        )

        # Phase 3: model-level aggregation
        total_supply = total_anticipated = total_demand = 0.0
        for vc in variant_capacities:
            total_supply += vc.total_capacity
            total_demand += vc.total_demand
            total_anticipated += (
                (vc.replica_count + vc.pending_replicas) * vc.per_replica_capacity
            )
        total_demand += estimate_scheduler_queue_demand(
            input.scheduler_queue,
            input.replica_metrics,
            getattr(cfg, "scheduler_queue_drain_factor", 1.0),
        )

        utilization = total_demand / total_supply if total_supply > 0 else 0.0

        # Phase 4: scaling signals
        required = 0.0
        if cfg.scale_up_threshold > 0:
            required = total_demand / cfg.scale_up_threshold - total_anticipated
        required = max(required, 0.0)
        spare = 0.0
        if cfg.scale_down_boundary > 0:
            spare = total_supply - total_demand / cfg.scale_down_boundary
        spare = max(spare, 0.0)

        return AnalyzerResult(
            analyzer_name=self.name(),
            model_id=input.model_id,
            namespace=input.namespace,
            analyzed_at=utcnow(),
            variant_capacities=variant_capacities,
            total_supply=total_supply,
            total_demand=total_demand,
            utilization=utilization,
            required_capacity=required,
            spare_capacity=spare,
        )

    # --- phases ---

    def _compute_replica_capacity(
        self,
        rm: ReplicaMetrics,
        cfg: SaturationScalingConfig,
        model_id: str,
        namespace: str,
        gpu_count: int,
    ) -> Optional[ReplicaCapacity]:
        if rm.total_kv_capacity_tokens <= 0:
            return None

        replica_demand = rm.tokens_in_use
        if rm.avg_input_tokens > 0:
            replica_demand += int(rm.queue_length) * int(rm.avg_input_tokens)

        k1 = int(rm.total_kv_capacity_tokens * cfg.kv_cache_threshold)

        vllm_params = None
        rec = self.capacity_store.get(namespace, model_id, rm.variant_name)
        if rec is not None:
            vllm_params = rec.vllm_params

        k2 = self._compute_k2(
            model_id,
            rm.accelerator_name,
            rm.queue_length,
            rm.tokens_in_use,
            rm.avg_output_tokens,
            rm.avg_input_tokens,
            cfg.queue_length_threshold,
            vllm_params,
            k1,
        )
        effective = min(k1, k2)
        is_saturated = replica_demand >= effective

        # Live data is authoritative; preserve parsed VLLMParams for
        # FindCompatible.
        existing_params = None
        existing = self.capacity_store.get(namespace, model_id, rm.variant_name)
        if existing is not None and existing.vllm_params is not None:
            existing_params = existing.vllm_params
        self.capacity_store.update(
            namespace,
            model_id,
            rm.variant_name,
            CapacityRecord(
                accelerator_name=rm.accelerator_name,
                gpu_count=gpu_count,
                num_gpu_blocks=rm.num_gpu_blocks,
                block_size=rm.block_size,
                total_kv_capacity_tokens=rm.total_kv_capacity_tokens,
                effective_capacity=effective,
                vllm_params=existing_params,
                learned_from="live",
            ),
        )

        return ReplicaCapacity(
            pod_name=rm.pod_name,
            variant_name=rm.variant_name,
            accelerator_name=rm.accelerator_name,
            tokens_in_use=rm.tokens_in_use,
            total_kv_capacity_tokens=rm.total_kv_capacity_tokens,
            memory_bound_capacity=k1,
            compute_bound_capacity=k2,
            effective_capacity=effective,
            is_saturated=is_saturated,
            replica_demand=replica_demand,
        )

    def _compute_k2(
        self,
        model_id: str,
        accelerator: str,
        queue_len: int,
        tokens_in_use: int,
        avg_output: float,
        avg_input: float,
        queue_threshold: float,
        vllm_params: Optional[VLLMEngineParams],
        k1: int,
    ) -> int:
        bucket = classify_output_length(avg_output)
        history_key = f"{model_id}|{accelerator}|{bucket}"

        # Priority 1: observed at queue saturation
        if queue_len >= int(queue_threshold) and tokens_in_use > 0:
            with self._lock:
                ra = self._compute_capacity_history.get(history_key)
                if ra is None:
                    ra = RollingAverage(ROLLING_AVERAGE_WINDOW_SIZE)
                    self._compute_capacity_history[history_key] = ra
                ra.add(float(tokens_in_use))
            return tokens_in_use

        # Priority 2: rolling history
        with self._lock:
            ra = self._compute_capacity_history.get(history_key)
            hist_avg = ra.average() if ra is not None else 0.0
        if hist_avg > 0:
            return int(hist_avg)

        # Priority 3: derived from deployment args
        derived = estimate_capacity_from_params(vllm_params, avg_input, avg_output)
        if derived > 0:
            return derived

        # Priority 4: fallback to k1
        return k1

    def _aggregate_by_variant(
        self,
        replica_capacities: List[ReplicaCapacity],
        input_metrics: List[ReplicaMetrics],
        variant_states: List[VariantReplicaState],
        model_id: str,
        namespace: str,
        kv_cache_threshold: float,
    ) -> List[VariantCapacity]:
        by_variant: Dict[str, List[ReplicaCapacity]] = {}
        for rc in replica_capacities:
            by_variant.setdefault(rc.variant_name, []).append(rc)

        variant_cost: Dict[str, float] = {}
        variant_accel: Dict[str, str] = {}
        for rm in input_metrics:
            if rm.variant_name not in variant_cost:
                variant_cost[rm.variant_name] = rm.cost
                variant_accel[rm.variant_name] = rm.accelerator_name

        model_avg_input, model_avg_output, _ = compute_model_workload_averages(
            input_metrics
        )

        result: List[VariantCapacity] = []
        for vs in variant_states:
            replicas = by_variant.get(vs.variant_name, [])
            per_replica = 0.0
            total_demand = 0.0
            accelerator = variant_accel.get(vs.variant_name, "")
            cost = variant_cost.get(vs.variant_name, 0.0)

            ready_count = max(vs.current_replicas - vs.pending_replicas, 0)

            if replicas:
                capacities = [rc.effective_capacity for rc in replicas]
                total_demand = float(sum(rc.replica_demand for rc in replicas))
                per_replica = float(_median(capacities))
                if not accelerator:
                    accelerator = replicas[0].accelerator_name
            else:
                rec = self.capacity_store.get(namespace, model_id, vs.variant_name)
                if rec is not None and rec.effective_capacity > 0:
                    per_replica = self._estimate_stored_capacity(
                        rec, model_id, kv_cache_threshold,
                        model_avg_input, model_avg_output,
                    )
                else:
                    compat = self._lookup_compatible_capacity(
                        namespace, model_id, vs.variant_name,
                        accelerator, vs.gpus_per_replica,
                    )
                    if compat is not None:
                        per_replica = float(compat.effective_capacity)

            total_capacity = ready_count * per_replica
            utilization = total_demand / total_capacity if total_capacity > 0 else 0.0

            result.append(
                VariantCapacity(
                    variant_name=vs.variant_name,
                    accelerator_name=accelerator,
                    cost=cost,
                    replica_count=ready_count,
                    pending_replicas=vs.pending_replicas,
                    per_replica_capacity=per_replica,
                    total_capacity=total_capacity,
                    total_demand=total_demand,
                    utilization=utilization,
                )
            )
        return result

    def _lookup_compatible_capacity(
        self,
        namespace: str,
        model_id: str,
        variant_name: str,
        accelerator: str,
        gpu_count: int,
    ) -> Optional[CapacityRecord]:
        rec = self.capacity_store.get(namespace, model_id, variant_name)
        if rec is None or rec.vllm_params is None:
            return None
        # Improvement over the reference: when no live metrics exist for the
        # variant the accelerator from the caller is empty; fall back to the
        # accelerator recorded when the deployment was parsed so the
        # cross-variant match can still succeed.
        if not accelerator:
            accelerator = rec.accelerator_name
        return self.capacity_store.find_compatible(
            model_id, accelerator, gpu_count, rec.vllm_params
        )

    def _estimate_stored_capacity(
        self,
        rec: CapacityRecord,
        model_id: str,
        kv_cache_threshold: float,
        model_avg_input: float,
        model_avg_output: float,
    ) -> float:
        if rec is None:
            return 0.0
        if rec.learned_from == "live":
            return float(rec.effective_capacity)
        if rec.vllm_params is not None and model_avg_output > 0:
            derived = estimate_capacity_from_params(
                rec.vllm_params, model_avg_input, model_avg_output
            )
            if derived > 0:
                bounded = derived
                if rec.total_kv_capacity_tokens > 0 and kv_cache_threshold > 0:
                    k1 = int(rec.total_kv_capacity_tokens * kv_cache_threshold)
                    if 0 < k1 < bounded:
                        bounded = k1
                compat = self.capacity_store.find_compatible(
                    model_id, rec.accelerator_name, rec.gpu_count, rec.vllm_params
                )
                if (
                    compat is not None
                    and compat.learned_from == "live"
                    and 0 < compat.effective_capacity < bounded
                ):
                    bounded = compat.effective_capacity
                return float(bounded)
        return float(rec.effective_capacity)
