"""Replica metrics collector — one Refresh for the 6 saturation queries,
merged into per-pod ReplicaMetrics.

Parity: reference internal/collector/replica_metrics.go:60-477 — pod label
fallback ("pod" → "pod_name"), NaN/Inf guards on rate-derived values,
prefix-hit-rate clamped to [0,1], pods with neither kv nor queue metrics
skipped, pods not matching any deployment skipped, accelerator from the VA
label `inference.optimization/acceleratorName`, cost from spec.variantCost,
TotalKvCapacityTokens = numGpuBlocks × blockSize (Python ints — no overflow
guard needed at MI355X scale), TokensInUse = round(kv × capacity) clamped.
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional

from ..analyzers.interfaces import (
    ReplicaMetrics,
    ReplicaMetricsMetadata,
    SchedulerQueueMetrics,
)
from ..api.types import VariantAutoscaling, utcnow
from ..constants import ACCELERATOR_LABEL_KEY, DEFAULT_VARIANT_COST
from ..kube.objects import Deployment
from ..utils.logging import get_logger
from . import registration as reg
from .pod_va_mapper import PodVAMapper
from .source import MetricsSource, RefreshSpec

log = get_logger("collector.replica_metrics")


class _PodMetricData:
    __slots__ = (
        "kv_usage", "has_kv", "queue_len", "has_queue",
        "num_gpu_blocks", "block_size", "has_cache_config",
        "avg_output_tokens", "avg_input_tokens", "prefix_cache_hit_rate",
    )

    def __init__(self):
        self.kv_usage = 0.0
        self.has_kv = False
        self.queue_len = 0
        self.has_queue = False
        self.num_gpu_blocks = 0
        self.block_size = 0
        self.has_cache_config = False
        self.avg_output_tokens = 0.0
        self.avg_input_tokens = 0.0
        self.prefix_cache_hit_rate = 0.0


def _pod_name(labels: Dict[str, str]) -> str:
    return labels.get("pod") or labels.get("pod_name") or ""


def _finite(v: float) -> bool:
    return not (math.isnan(v) or math.isinf(v))


class ReplicaMetricsCollector:
    def __init__(
        self,
        source: MetricsSource,
        pod_va_mapper: PodVAMapper,
        freshness=None,  # config.FreshnessThresholds (1m/2m/5m ladder)
    ):
        self.source = source
        self.pod_va_mapper = pod_va_mapper
        self.freshness = freshness

    def collect_replica_metrics(
        self,
        model_id: str,
        namespace: str,
        deployments: Dict[str, Deployment],
        variant_autoscalings: Dict[str, VariantAutoscaling],
        variant_costs: Dict[str, float],
    ) -> List[ReplicaMetrics]:
        params = {"modelID": model_id, "namespace": namespace}
        queries = [
            reg.QUERY_KV_CACHE_USAGE,
            reg.QUERY_QUEUE_LENGTH,
            reg.QUERY_CACHE_CONFIG_INFO,
            reg.QUERY_AVG_OUTPUT_TOKENS,
            reg.QUERY_AVG_INPUT_TOKENS,
            reg.QUERY_PREFIX_CACHE_HIT_RATE,
        ]
        results = self.source.refresh(RefreshSpec(queries=queries, params=params))

        pod_data: Dict[str, _PodMetricData] = {}

        def data_for(labels: Dict[str, str]) -> Optional[_PodMetricData]:
            name = _pod_name(labels)
            if not name:
                return None
            return pod_data.setdefault(name, _PodMetricData())

        sample_ages: Dict[str, float] = {}

        kv = results.get(reg.QUERY_KV_CACHE_USAGE)
        if kv is not None:
            if kv.has_error():
                raise RuntimeError(f"KV cache query failed: {kv.error}")
            for v in kv.values:
                d = data_for(v.labels)
                # NaN guard: the exposition format allows NaN samples and
                # only PrometheusSource maps them to 0 — int()/round() on
                # NaN would crash the whole collection
                if d is not None and _finite(v.value):
                    d.kv_usage = v.value
                    d.has_kv = True
                    name = _pod_name(v.labels)
                    if name:
                        sample_ages[name] = v.age_seconds()

        q = results.get(reg.QUERY_QUEUE_LENGTH)
        if q is not None:
            if q.has_error():
                raise RuntimeError(f"queue length query failed: {q.error}")
            for v in q.values:
                d = data_for(v.labels)
                if d is not None and _finite(v.value):
                    d.queue_len = int(v.value)
                    d.has_queue = True

        cc = results.get(reg.QUERY_CACHE_CONFIG_INFO)
        if cc is not None and not cc.has_error():
            for v in cc.values:
                d = data_for(v.labels)
                if d is None:
                    continue
                try:
                    blocks = int(v.labels.get("num_gpu_blocks", "") or 0)
                except ValueError:
                    blocks = 0
                try:
                    size = int(v.labels.get("block_size", "") or 0)
                except ValueError:
                    size = 0
                if blocks:
                    d.num_gpu_blocks = blocks
                if size:
                    d.block_size = size
                if d.num_gpu_blocks > 0 and d.block_size > 0:
                    d.has_cache_config = True

        for query, attr in (
            (reg.QUERY_AVG_OUTPUT_TOKENS, "avg_output_tokens"),
            (reg.QUERY_AVG_INPUT_TOKENS, "avg_input_tokens"),
        ):
            res = results.get(query)
            if res is not None and not res.has_error():
                for v in res.values:
                    d = data_for(v.labels)
                    if d is not None and _finite(v.value):
                        setattr(d, attr, v.value)

        hit = results.get(reg.QUERY_PREFIX_CACHE_HIT_RATE)
        if hit is not None and not hit.has_error():
            for v in hit.values:
                d = data_for(v.labels)
                if d is not None and _finite(v.value) and 0 <= v.value <= 1:
                    d.prefix_cache_hit_rate = v.value

        # Build ReplicaMetrics
        replica_metrics: List[ReplicaMetrics] = []
        collected_at = utcnow()
        for pod_name in sorted(pod_data):
            data = pod_data[pod_name]
            if not data.has_kv and not data.has_queue:
                continue
            kv_usage = data.kv_usage if data.has_kv else 0.0
            queue_len = data.queue_len if data.has_queue else 0

            va_name = self.pod_va_mapper.find_va_for_pod(
                pod_name, namespace, deployments
            )
            if not va_name:
                log.info(
                    "skipping pod %s: matches no deployment of model %s",
                    pod_name,
                    model_id,
                )
                continue
            variant_key = f"{namespace}/{va_name}"

            accelerator_name = ""
            va = variant_autoscalings.get(variant_key)
            if va is not None:
                accelerator_name = va.metadata.labels.get(ACCELERATOR_LABEL_KEY, "")

            cost = variant_costs.get(variant_key, DEFAULT_VARIANT_COST)

            total_kv_capacity = 0
            tokens_in_use = 0
            if data.has_cache_config:
                total_kv_capacity = data.num_gpu_blocks * data.block_size
                rounded = round(kv_usage * total_kv_capacity)
                tokens_in_use = int(min(max(rounded, 0), total_kv_capacity))

            replica_metrics.append(
                ReplicaMetrics(
                    pod_name=pod_name,
                    model_id=model_id,
                    namespace=namespace,
                    variant_name=va_name,
                    accelerator_name=accelerator_name,
                    kv_cache_usage=kv_usage,
                    queue_length=queue_len,
                    cost=cost,
                    num_gpu_blocks=data.num_gpu_blocks,
                    block_size=data.block_size,
                    total_kv_capacity_tokens=total_kv_capacity,
                    tokens_in_use=tokens_in_use,
                    avg_output_tokens=data.avg_output_tokens,
                    avg_input_tokens=data.avg_input_tokens,
                    prefix_cache_hit_rate=data.prefix_cache_hit_rate,
                    metadata=ReplicaMetricsMetadata(
                        collected_at=collected_at,
                        age_seconds=sample_ages.get(pod_name, 0.0),
                        freshness_status=(
                            self.freshness.determine_status(
                                sample_ages.get(pod_name, 0.0)
                            )
                            if self.freshness is not None
                            else "fresh"
                        ),
                    ),
                )
            )
        return replica_metrics

    def collect_scheduler_queue_metrics(
        self, model_id: str
    ) -> Optional[SchedulerQueueMetrics]:
        """Model-level EPP flow-control queue metrics; None when unavailable."""
        try:
            results = self.source.refresh(
                RefreshSpec(
                    queries=[
                        reg.QUERY_SCHEDULER_QUEUE_SIZE,
                        reg.QUERY_SCHEDULER_QUEUE_BYTES,
                    ],
                    params={"modelID": model_id},
                )
            )
        except Exception as e:  # noqa: BLE001
            log.debug("scheduler queue metrics unavailable for %s: %s", model_id, e)
            return None

        queue_size = queue_bytes = 0
        has_data = False
        res = results.get(reg.QUERY_SCHEDULER_QUEUE_SIZE)
        if res is not None and not res.has_error():
            for v in res.values:
                if _finite(v.value):
                    queue_size += int(v.value)
                    has_data = True
        res = results.get(reg.QUERY_SCHEDULER_QUEUE_BYTES)
        if res is not None and not res.has_error():
            for v in res.values:
                if _finite(v.value):
                    queue_bytes += int(v.value)
                    has_data = True
        if not has_data:
            return None
        return SchedulerQueueMetrics(queue_size=queue_size, queue_bytes=queue_bytes)
