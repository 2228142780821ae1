"""Limiter interfaces + DefaultLimiter.

Parity: reference internal/engines/pipeline/limiter_interfaces.go:72-222 and
default_limiter.go:42-140. Separation of concerns: the Inventory owns
granularity (per-type pools), the AllocationAlgorithm owns strategy, the
DefaultLimiter wires them and tracks decision metadata. The V2 path
(compute_constraints) exposes pools rather than mutating decisions.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Protocol

from ..analyzers.interfaces import AnalyzerResult, VariantDecision, VariantReplicaState


@dataclass
class ResourcePool:
    limit: int = 0
    used: int = 0
    available: int = 0


@dataclass
class ResourceConstraints:
    provider_name: str = ""
    pools: Dict[str, ResourcePool] = field(default_factory=dict)
    total_limit: int = 0
    total_used: int = 0
    total_avail: int = 0


@dataclass
class ModelScalingRequest:
    """Optimizer input (optimizer_interfaces.go:11)."""

    model_id: str = ""
    namespace: str = ""
    result: Optional[AnalyzerResult] = None
    variant_states: List[VariantReplicaState] = field(default_factory=list)


class ResourceAllocator(Protocol):
    def try_allocate(self, decision: VariantDecision, gpus_requested: int) -> int: ...
    def remaining(self) -> int: ...


class AllocationAlgorithm(Protocol):
    def name(self) -> str: ...
    def allocate(
        self, decisions: List[VariantDecision], allocator: ResourceAllocator
    ) -> None: ...


class Inventory(Protocol):
    def refresh(self) -> None: ...
    def set_used(self, used_by_type: Dict[str, int]) -> None: ...
    def create_allocator(self) -> ResourceAllocator: ...
    def total_limit(self) -> int: ...
    def total_used(self) -> int: ...
    def total_available(self) -> int: ...
    def get_resource_pools(self) -> Dict[str, ResourcePool]: ...


class DefaultLimiter:
    """Inventory + algorithm → constrained decisions."""

    def __init__(self, name: str, inventory: Inventory, algorithm: AllocationAlgorithm):
        self._name = name
        self.inventory = inventory
        self.algorithm = algorithm

    def name(self) -> str:
        return self._name

    def limit(self, decisions: List[VariantDecision]) -> None:
        """V1 path: modify decisions in place based on available GPUs."""
        if not decisions:
            return
        for d in decisions:
            # preserve the pre-limit desire for observability (the V1
            # target builder sets this; the V2 optimizer's decisions
            # arrive with the field unset)
            if d.original_target_replicas <= 0:
                d.original_target_replicas = d.target_replicas
        self.inventory.refresh()
        self.inventory.set_used(self._calculate_used_gpus(decisions))
        allocator = self.inventory.create_allocator()
        self.algorithm.allocate(decisions, allocator)
        self._update_decision_metadata(decisions)

    @staticmethod
    def _calculate_used_gpus(decisions: List[VariantDecision]) -> Dict[str, int]:
        used: Dict[str, int] = {}
        for d in decisions:
            if not d.accelerator_name:
                continue
            used[d.accelerator_name] = (
                used.get(d.accelerator_name, 0)
                + d.current_replicas * d.gpus_per_replica
            )
        return used

    def _update_decision_metadata(self, decisions: List[VariantDecision]) -> None:
        for d in decisions:
            if d.was_limited:
                d.limited_by = self._name
            d.add_decision_step(self._name, self._build_step_reason(d), d.was_limited)

    @staticmethod
    def _build_step_reason(d: VariantDecision) -> str:
        change = d.target_replicas - d.current_replicas
        if change <= 0:
            if d.was_limited:
                # the ask was truncated all the way back to current —
                # say so instead of a misleading "no scale-up"
                return (
                    f"limited: pool exhausted, 0 GPUs allocated (wanted "
                    f"{d.original_target_replicas}, kept "
                    f"{d.target_replicas})"
                )
            return f"no scale-up (target={d.target_replicas}, current={d.current_replicas})"
        if d.was_limited:
            return f"limited: allocated {d.gpus_allocated} GPUs for +{change} replicas"
        return f"allocated {d.gpus_allocated} GPUs for +{change} replicas"

    def compute_constraints(
        self, current_usage: Dict[str, int]
    ) -> ResourceConstraints:
        """V2 path: expose pools instead of mutating decisions."""
        self.inventory.refresh()
        self.inventory.set_used(current_usage)
        return ResourceConstraints(
            provider_name=self._name,
            pools=self.inventory.get_resource_pools(),
            total_limit=self.inventory.total_limit(),
            total_used=self.inventory.total_used(),
            total_avail=self.inventory.total_available(),
        )
