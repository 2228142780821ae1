"""Shared analyzer / decision-pipeline data types.

Parity: reference internal/interfaces/saturation_analyzer.go:12-243 and
internal/interfaces/analyzer.go:15-113. Same fields and semantics; Python
dataclasses with snake_case names.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from datetime import datetime
from typing import List, Optional

from ..api.types import CrossVersionObjectReference, utcnow

# SaturationAction values (saturation_analyzer.go:219-225)
ACTION_SCALE_UP = "scale-up"
ACTION_SCALE_DOWN = "scale-down"
ACTION_NONE = "none"


@dataclass
class ReplicaMetricsMetadata:
    """Freshness info: collected_at, age (seconds), status fresh|stale|unavailable."""

    collected_at: Optional[datetime] = None
    age_seconds: float = 0.0
    freshness_status: str = "fresh"


@dataclass
class ReplicaMetrics:
    """Capacity metrics for a single vLLM replica (pod).

    V2 fields: on MI355X TotalKvCapacityTokens routinely reaches ~2.3x the
    MI300X value (288 GB HBM3E); all token math is plain Python ints, so the
    reference's int64 overflow guard (replica_metrics.go:356-359) is not
    needed — validated in tests at 288 GB-scale block counts.
    """

    pod_name: str = ""
    kv_cache_usage: float = 0.0  # 0.0-1.0
    queue_length: int = 0
    variant_name: str = ""
    namespace: str = ""
    model_id: str = ""
    accelerator_name: str = ""
    cost: float = 10.0
    metadata: Optional[ReplicaMetricsMetadata] = None

    # --- V2 token-capacity fields ---
    num_gpu_blocks: int = 0
    block_size: int = 0
    total_kv_capacity_tokens: int = 0
    tokens_in_use: int = 0
    avg_output_tokens: float = 0.0
    avg_input_tokens: float = 0.0
    prefix_cache_hit_rate: float = 0.0


@dataclass
class SchedulerQueueMetrics:
    """Model-level flow-control queue metrics from the llm-d EPP scheduler."""

    queue_size: int = 0
    queue_bytes: int = 0


@dataclass
class VariantReplicaState:
    """Current/desired/pending replica counts + GPUs per replica for a variant."""

    variant_name: str = ""
    current_replicas: int = 0
    desired_replicas: int = 0  # from optimizer/CRD status, 0 if unset
    pending_replicas: int = 0  # current - ready (anti-cascade signal)
    gpus_per_replica: int = 1  # from amd.com/gpu (or nvidia/intel) requests


@dataclass
class VariantSaturationAnalysis:
    variant_name: str = ""
    accelerator_name: str = ""
    cost: float = 10.0
    replica_count: int = 0
    non_saturated_count: int = 0
    max_kv_cache_usage: float = 0.0
    max_queue_length: int = 0
    avg_spare_kv_capacity: float = 0.0
    avg_spare_queue_length: float = 0.0
    saturated_replicas: List[str] = field(default_factory=list)


@dataclass
class ModelSaturationAnalysis:
    model_id: str = ""
    namespace: str = ""
    analyzed_at: Optional[datetime] = None
    total_replicas: int = 0
    non_saturated_count: int = 0
    avg_spare_kv_capacity: float = 0.0
    avg_spare_queue_length: float = 0.0
    should_scale_up: bool = False
    scale_up_reason: str = ""
    scale_down_safe: bool = False
    variant_analyses: List[VariantSaturationAnalysis] = field(default_factory=list)


@dataclass
class AnalyzerInput:
    """Common input to all analyzers (interfaces/analyzer.go:32-45)."""

    model_id: str = ""
    namespace: str = ""
    replica_metrics: List[ReplicaMetrics] = field(default_factory=list)
    variant_states: List[VariantReplicaState] = field(default_factory=list)
    config: object = None  # AnalyzerConfig (SaturationScalingConfig)
    scheduler_queue: Optional[SchedulerQueueMetrics] = None


@dataclass
class VariantCapacity:
    """Per-variant capacity in analyzer-specific units (tokens for V2)."""

    variant_name: str = ""
    accelerator_name: str = ""
    cost: float = 10.0
    replica_count: int = 0
    pending_replicas: int = 0
    per_replica_capacity: float = 0.0
    total_capacity: float = 0.0
    total_demand: float = 0.0
    utilization: float = 0.0


@dataclass
class AnalyzerResult:
    """Common analyzer output (interfaces/analyzer.go:70-95)."""

    analyzer_name: str = ""
    model_id: str = ""
    namespace: str = ""
    analyzed_at: Optional[datetime] = None
    variant_capacities: List[VariantCapacity] = field(default_factory=list)
    total_supply: float = 0.0
    total_demand: float = 0.0
    utilization: float = 0.0
    required_capacity: float = 0.0  # >0 → scale-up needed
    spare_capacity: float = 0.0  # >0 → scale-down possible


@dataclass
class DecisionStep:
    """One pipeline stage's contribution to a decision."""

    name: str = ""
    action: str = ACTION_NONE
    target_replicas: int = 0
    reason: str = ""
    was_constrained: bool = False
    timestamp: Optional[datetime] = None


@dataclass
class VariantDecision:
    """Shared scaling-decision state flowing through the pipeline
    (interfaces/saturation_analyzer.go:126-216)."""

    variant_name: str = ""
    namespace: str = ""
    model_id: str = ""
    accelerator_name: str = ""
    cost: float = 10.0

    action: str = ACTION_NONE
    current_replicas: int = 0
    target_replicas: int = 0
    original_target_replicas: int = 0
    desired_replicas: int = 0

    gpus_per_replica: int = 1
    spare_capacity: float = 0.0
    scale_target_ref: Optional[CrossVersionObjectReference] = None

    decision_steps: List[DecisionStep] = field(default_factory=list)
    reason: str = ""

    saturation_based: bool = True
    model_based_decision: bool = False
    safety_override: bool = False
    last_run_time: Optional[datetime] = None
    saturation_only: bool = True

    gpus_allocated: int = 0
    was_limited: bool = False
    limited_by: str = ""

    metrics_available: bool = True
    metrics_reason: str = ""
    metrics_message: str = ""

    # OptimizationReady condition payload, carried through the DecisionCache
    # so the reconciler (the single status writer) can persist the condition
    # the engine computed. The reference sets this condition only on the
    # engine's local copy (engine.go:915-950) and it is never persisted;
    # carrying it here is a documented improvement.
    optimization_ready_reason: str = ""
    optimization_ready_message: str = ""

    def add_decision_step(self, name: str, reason: str, was_constrained: bool) -> None:
        self.decision_steps.append(
            DecisionStep(
                name=name,
                action=self.action,
                target_replicas=self.target_replicas,
                reason=reason,
                was_constrained=was_constrained,
                timestamp=utcnow(),
            )
        )
        self.reason = reason
