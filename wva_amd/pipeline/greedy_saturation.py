"""Greedy-by-saturation allocation algorithm.

Parity: reference internal/engines/pipeline/greedy_saturation_algorithm.go
:34-106 — scale-up candidates sorted by spare_capacity ascending (most
saturated first), cost ascending as tie-break; partial allocation truncates
target_replicas to whole replicas and sets was_limited.
"""
from __future__ import annotations

from typing import List

from ..analyzers.interfaces import VariantDecision
from .limiter import ResourceAllocator


class GreedyBySaturation:
    def name(self) -> str:
        return "greedy-by-saturation"

    def allocate(
        self, decisions: List[VariantDecision], allocator: ResourceAllocator
    ) -> None:
        candidates = [
            d for d in decisions if d.target_replicas > d.current_replicas
        ]
        candidates.sort(key=lambda d: (d.spare_capacity, d.cost))
        for d in candidates:
            self._allocate_for_decision(d, allocator)

    @staticmethod
    def _allocate_for_decision(
        d: VariantDecision, allocator: ResourceAllocator
    ) -> None:
        replicas_needed = d.target_replicas - d.current_replicas
        if replicas_needed <= 0:
            return
        gpus_per_replica = d.gpus_per_replica if d.gpus_per_replica > 0 else 1
        gpus_requested = replicas_needed * gpus_per_replica
        gpus_allocated = allocator.try_allocate(d, gpus_requested)
        replicas_allocated = gpus_allocated // gpus_per_replica
        d.gpus_allocated = replicas_allocated * gpus_per_replica
        d.target_replicas = d.current_replicas + replicas_allocated
        if replicas_allocated < replicas_needed:
            d.was_limited = True
