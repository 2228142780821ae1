"""V1 percentage-based saturation analyzer.

Parity: reference internal/saturation/analyzer.go:31-439 and
internal/saturation/constants.go. Behavior is identical:

  * saturated  iff kv_usage >= kvCacheThreshold OR queue >= queueLengthThreshold
  * spare      = threshold - usage, averaged over non-saturated replicas
  * scale-up   iff avgSpareKv < kvSpareTrigger OR avgSpareQueue < queueSpareTrigger
  * scale-down safe iff >= 2 non-saturated replicas AND the N/(N-1)
    load-redistribution simulation keeps spare >= triggers
  * target calc: transition freeze; +1 on the cheapest variant without
    pending replicas (name-ascending tie-break); -1 on the most expensive
    (floor 1, name-descending tie-break)
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from ..config.saturation import SaturationScalingConfig
from ..utils.logging import get_logger
from .interfaces import (
    ModelSaturationAnalysis,
    ReplicaMetrics,
    VariantReplicaState,
    VariantSaturationAnalysis,
)
from ..api.types import utcnow

MIN_NON_SATURATED_REPLICAS_FOR_SCALE_DOWN = 2

log = get_logger("saturation.v1")


class SaturationAnalyzerV1:
    """Percentage-threshold saturation analyzer (the default path)."""

    def analyze_model_saturation(
        self,
        model_id: str,
        namespace: str,
        replica_metrics: List[ReplicaMetrics],
        config: SaturationScalingConfig,
    ) -> ModelSaturationAnalysis:
        if not replica_metrics:
            return ModelSaturationAnalysis(
                model_id=model_id,
                namespace=namespace,
                analyzed_at=utcnow(),
                total_replicas=0,
                should_scale_up=False,
                scale_down_safe=False,
                variant_analyses=[],
            )

        analysis = ModelSaturationAnalysis(
            model_id=model_id, namespace=namespace, analyzed_at=utcnow()
        )

        variant_map: Dict[str, List[ReplicaMetrics]] = {}
        for m in replica_metrics:
            variant_map.setdefault(m.variant_name, []).append(m)

        total_spare_kv = 0.0
        total_spare_queue = 0.0
        non_saturated = 0
        variant_analyses: List[VariantSaturationAnalysis] = []
        for variant_name, metrics in variant_map.items():
            va = self._analyze_variant(variant_name, metrics, config)
            variant_analyses.append(va)
            non_saturated += va.non_saturated_count
            total_spare_kv += va.avg_spare_kv_capacity * va.non_saturated_count
            total_spare_queue += va.avg_spare_queue_length * va.non_saturated_count

        analysis.total_replicas = len(replica_metrics)
        analysis.non_saturated_count = non_saturated
        analysis.variant_analyses = variant_analyses

        if non_saturated > 0:
            analysis.avg_spare_kv_capacity = total_spare_kv / non_saturated
            analysis.avg_spare_queue_length = total_spare_queue / non_saturated

        analysis.should_scale_up, analysis.scale_up_reason = self._should_scale_up(
            analysis.avg_spare_kv_capacity, analysis.avg_spare_queue_length, config
        )
        analysis.scale_down_safe = self._is_scale_down_safe(
            non_saturated,
            analysis.avg_spare_kv_capacity,
            analysis.avg_spare_queue_length,
            config,
        )
        return analysis

    def _analyze_variant(
        self,
        variant_name: str,
        metrics: List[ReplicaMetrics],
        config: SaturationScalingConfig,
    ) -> VariantSaturationAnalysis:
        va = VariantSaturationAnalysis(
            variant_name=variant_name,
            replica_count=len(metrics),
            saturated_replicas=[],
        )
        if metrics:
            va.accelerator_name = metrics[0].accelerator_name
            va.cost = metrics[0].cost

        total_spare_kv = 0.0
        total_spare_queue = 0.0
        non_saturated = 0
        for m in metrics:
            is_saturated = (
                m.kv_cache_usage >= config.kv_cache_threshold
                or float(m.queue_length) >= config.queue_length_threshold
            )
            if is_saturated:
                va.saturated_replicas.append(m.pod_name)
            else:
                total_spare_kv += config.kv_cache_threshold - m.kv_cache_usage
                total_spare_queue += config.queue_length_threshold - float(
                    m.queue_length
                )
                non_saturated += 1
            va.max_kv_cache_usage = max(va.max_kv_cache_usage, m.kv_cache_usage)
            va.max_queue_length = max(va.max_queue_length, m.queue_length)

        va.non_saturated_count = non_saturated
        if non_saturated > 0:
            va.avg_spare_kv_capacity = total_spare_kv / non_saturated
            va.avg_spare_queue_length = total_spare_queue / non_saturated
        return va

    def _should_scale_up(
        self,
        avg_spare_kv: float,
        avg_spare_queue: float,
        config: SaturationScalingConfig,
    ) -> Tuple[bool, str]:
        kv_triggered = avg_spare_kv < config.kv_spare_trigger
        queue_triggered = avg_spare_queue < config.queue_spare_trigger
        if not kv_triggered and not queue_triggered:
            return False, ""
        if kv_triggered and queue_triggered:
            return True, (
                f"both KV spare ({avg_spare_kv:.3f} < {config.kv_spare_trigger:.3f}) "
                f"and queue spare ({avg_spare_queue:.1f} < {config.queue_spare_trigger:.1f})"
            )
        if kv_triggered:
            return True, (
                f"KV spare capacity low ({avg_spare_kv:.3f} < {config.kv_spare_trigger:.3f})"
            )
        return True, (
            f"queue spare capacity low ({avg_spare_queue:.1f} < {config.queue_spare_trigger:.1f})"
        )

    def _is_scale_down_safe(
        self,
        non_saturated_count: int,
        avg_spare_kv: float,
        avg_spare_queue: float,
        config: SaturationScalingConfig,
    ) -> bool:
        if non_saturated_count < MIN_NON_SATURATED_REPLICAS_FOR_SCALE_DOWN:
            return False
        avg_kv_load = config.kv_cache_threshold - avg_spare_kv
        avg_queue_load = config.queue_length_threshold - avg_spare_queue
        scale_factor = non_saturated_count / (non_saturated_count - 1)
        remaining_spare_kv = config.kv_cache_threshold - avg_kv_load * scale_factor
        remaining_spare_queue = (
            config.queue_length_threshold - avg_queue_load * scale_factor
        )
        return (
            remaining_spare_kv >= config.kv_spare_trigger
            and remaining_spare_queue >= config.queue_spare_trigger
        )

    def calculate_saturation_targets(
        self,
        saturation_analysis: Optional[ModelSaturationAnalysis],
        variant_states: List[VariantReplicaState],
    ) -> Dict[str, int]:
        """Per-variant target replicas. See module docstring for the rules."""
        targets: Dict[str, int] = {}
        if saturation_analysis is None or not saturation_analysis.variant_analyses:
            for state in variant_states:
                targets[state.variant_name] = state.current_replicas
            return targets

        state_map = {s.variant_name: s for s in variant_states}

        def state_of(name: str) -> VariantReplicaState:
            return state_map.get(name, VariantReplicaState(variant_name=name))

        # STEP 1: model-level transition detection
        model_in_transition = False
        transition_reasons: List[str] = []
        for va in saturation_analysis.variant_analyses:
            state = state_of(va.variant_name)
            if state.desired_replicas != 0 and (
                state.desired_replicas != state.current_replicas
            ):
                model_in_transition = True
                transition_reasons.append(
                    f"{va.variant_name}: desired({state.desired_replicas})"
                    f"!=current({state.current_replicas})"
                )
            if va.replica_count != state.current_replicas:
                model_in_transition = True
                transition_reasons.append(
                    f"{va.variant_name}: metrics({va.replica_count})"
                    f"!=current({state.current_replicas})"
                )

        # STEP 2: initialize targets
        for va in saturation_analysis.variant_analyses:
            state = state_of(va.variant_name)
            if model_in_transition:
                if state.desired_replicas != 0 and (
                    state.desired_replicas != state.current_replicas
                ):
                    targets[va.variant_name] = state.desired_replicas
                else:
                    targets[va.variant_name] = state.current_replicas
            else:
                targets[va.variant_name] = va.replica_count

        # STEP 3: transition freeze
        if model_in_transition:
            log.info(
                "model in transition, blocking scaling decisions: model=%s reasons=%s",
                saturation_analysis.model_id,
                transition_reasons,
            )
            return targets

        # STEP 4: stable — scale decisions
        if saturation_analysis.should_scale_up:
            cheapest: Optional[VariantSaturationAnalysis] = None
            for va in saturation_analysis.variant_analyses:
                if state_of(va.variant_name).pending_replicas > 0:
                    continue
                if (
                    cheapest is None
                    or va.cost < cheapest.cost
                    or (va.cost == cheapest.cost and va.variant_name < cheapest.variant_name)
                ):
                    cheapest = va
            if cheapest is not None:
                targets[cheapest.variant_name] += 1
        elif saturation_analysis.scale_down_safe:
            most_expensive: Optional[VariantSaturationAnalysis] = None
            for va in saturation_analysis.variant_analyses:
                if targets[va.variant_name] <= 1:
                    continue
                if (
                    most_expensive is None
                    or va.cost > most_expensive.cost
                    or (
                        va.cost == most_expensive.cost
                        and va.variant_name > most_expensive.variant_name
                    )
                ):
                    most_expensive = va
            if most_expensive is not None:
                targets[most_expensive.variant_name] -= 1

        return targets
