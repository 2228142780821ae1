"""Cost-aware scaling optimizer (V2 unlimited mode).

Parity: reference internal/engines/pipeline/cost_aware_optimizer.go:39-296 —
scale-up fills required_capacity by cost-efficiency (cost/perReplicaCapacity)
ascending with ceil(remaining/perReplicaCapacity) replicas; scale-down
removes floor(remaining/perReplicaCapacity) from highest-cost variants while
spare remains; the cheapest variant is protected at 1 replica only when it
is the last variant with replicas.
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional

from ..analyzers.interfaces import (
    ACTION_SCALE_DOWN,
    ACTION_SCALE_UP,
    AnalyzerResult,
    VariantCapacity,
    VariantDecision,
    VariantReplicaState,
)
from .limiter import ModelScalingRequest, ResourceConstraints

ACTION_NO_CHANGE = "no-change"

_INF = float("inf")


def _cost_efficiency(vc: VariantCapacity) -> float:
    if vc.per_replica_capacity <= 0:
        return _INF
    return vc.cost / vc.per_replica_capacity


class CostAwareOptimizer:
    def name(self) -> str:
        return "cost-aware"

    def optimize(
        self,
        requests: List[ModelScalingRequest],
        constraints: Optional[List[ResourceConstraints]] = None,
    ) -> List[VariantDecision]:
        all_decisions: List[VariantDecision] = []
        for req in requests:
            if req.result is None:
                continue
            state_map = {s.variant_name: s for s in req.variant_states}
            vc_map = {vc.variant_name: vc for vc in req.result.variant_capacities}
            targets = {s.variant_name: s.current_replicas for s in req.variant_states}

            if req.result.required_capacity > 0:
                self._scale_up(req.result, targets)
            elif req.result.spare_capacity > 0:
                self._scale_down(req.result, targets)

            all_decisions.extend(
                self._build_decisions(req, state_map, vc_map, targets)
            )
        return all_decisions

    @staticmethod
    def _scale_up(result: AnalyzerResult, targets: Dict[str, int]) -> None:
        sorted_vcs = sorted(result.variant_capacities, key=_cost_efficiency)
        remaining = result.required_capacity
        for vc in sorted_vcs:
            if remaining <= 0:
                break
            if vc.per_replica_capacity <= 0:
                continue
            needed = math.ceil(remaining / vc.per_replica_capacity)
            targets[vc.variant_name] = targets.get(vc.variant_name, 0) + needed
            remaining -= needed * vc.per_replica_capacity

    @staticmethod
    def _scale_down(result: AnalyzerResult, targets: Dict[str, int]) -> None:
        sorted_vcs = sorted(
            result.variant_capacities, key=lambda vc: vc.cost, reverse=True
        )
        remaining = result.spare_capacity
        for vc in sorted_vcs:
            if remaining <= 0:
                break
            if vc.per_replica_capacity <= 0:
                continue
            current = targets.get(vc.variant_name, 0)
            # spare-driven scale-down never drains the model to zero —
            # that transition belongs to the enforcer's scale-to-zero
            # policy. Removal order is most-expensive-first, so the
            # floor lands on the cheapest variant still holding
            # replicas. (Protecting a precomputed "cheapest" NAME is
            # wrong when the cheapest variant has no replicas — e.g.
            # cost ties — and would let the last real holder drain.)
            other_has = any(
                t > 0 for name, t in targets.items()
                if name != vc.variant_name
            )
            min_replicas = 0 if other_has else 1
            removable = current - min_replicas
            if removable <= 0:
                continue
            to_remove = min(
                int(math.floor(remaining / vc.per_replica_capacity)), removable
            )
            if to_remove <= 0:
                continue
            targets[vc.variant_name] = current - to_remove
            remaining -= to_remove * vc.per_replica_capacity

    @staticmethod
    def _build_decisions(
        req: ModelScalingRequest,
        state_map: Dict[str, VariantReplicaState],
        vc_map: Dict[str, VariantCapacity],
        targets: Dict[str, int],
    ) -> List[VariantDecision]:
        decisions = []
        for name, target in targets.items():
            state = state_map.get(name, VariantReplicaState(variant_name=name))
            vc = vc_map.get(name, VariantCapacity(variant_name=name))
            if target > state.current_replicas:
                action = ACTION_SCALE_UP
                reason = (
                    f"V2 scale-up (optimizer: cost-aware, "
                    f"required: {req.result.required_capacity:.0f})"
                )
            elif target < state.current_replicas:
                action = ACTION_SCALE_DOWN
                reason = (
                    f"V2 scale-down (optimizer: cost-aware, "
                    f"spare: {req.result.spare_capacity:.0f})"
                )
            else:
                action = ACTION_NO_CHANGE
                reason = "V2 steady state"
            decisions.append(
                VariantDecision(
                    variant_name=name,
                    model_id=req.model_id,
                    namespace=req.namespace,
                    accelerator_name=vc.accelerator_name,
                    cost=vc.cost,
                    current_replicas=state.current_replicas,
                    target_replicas=target,
                    gpus_per_replica=state.gpus_per_replica,
                    action=action,
                    reason=reason,
                )
            )
        return decisions
