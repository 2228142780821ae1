"""Per-accelerator-type GPU inventory.

Parity: reference internal/engines/pipeline/type_inventory.go:82-383 —
separate pools per accelerator type (MI355X workloads never consume MI300X
GPUs), normalized product names, non-thread-safe per-batch allocator.

MI355X addition: hive_limit_by_type() (from discovery.max_hive_by_type) lets
the allocator reject replicas whose gpus_per_replica exceeds any single
node's pool — an 8-GPU TP replica must land inside one xGMI hive.
"""
from __future__ import annotations

import threading
from typing import Dict, Optional

from ..analyzers.interfaces import VariantDecision
from ..discovery.gpu_operator import (
    K8sGpuOperatorDiscovery,
    normalize_accelerator_name,
)
from .limiter import ResourcePool


class TypeAllocator:
    """Per-type GPU allocator. NOT thread-safe; create per decision batch."""

    def __init__(
        self,
        remaining_by_type: Dict[str, int],
        hive_by_type: Optional[Dict[str, int]] = None,
    ):
        self._remaining = dict(remaining_by_type)
        self._total = sum(remaining_by_type.values())
        self._hive_by_type = dict(hive_by_type or {})

    def try_allocate(self, decision: VariantDecision, gpus_requested: int) -> int:
        if gpus_requested <= 0:
            return 0
        acc_type = decision.accelerator_name
        if not acc_type:
            raise ValueError(
                f"decision for {decision.namespace}/{decision.variant_name} "
                "has no accelerator_name"
            )
        # xGMI hive feasibility: a replica spanning more GPUs than any
        # single node of this type offers can never be scheduled.
        if self._hive_by_type:
            hive = self._hive_by_type.get(acc_type, 0)
            if 0 < hive < max(decision.gpus_per_replica, 1):
                return 0
        available = self._remaining.get(acc_type, 0)
        if available <= 0:
            return 0
        allocated = min(gpus_requested, available)
        # whole-replica granularity: a partial replica's GPUs would be
        # deducted from the pool but unusable (the algorithm truncates to
        # whole replicas), silently starving later candidates in the
        # same batch
        gpr = max(decision.gpus_per_replica, 1)
        allocated -= allocated % gpr
        self._remaining[acc_type] = available - allocated
        self._total -= allocated
        return allocated

    def remaining(self) -> int:
        return self._total

    def remaining_for_type(self, acc_type: str) -> int:
        return self._remaining.get(acc_type, 0)


class TypeInventory:
    def __init__(
        self,
        name: str,
        discovery: K8sGpuOperatorDiscovery,
        with_usage: bool = False,
        hive_aware: bool = True,
    ):
        self._name = name
        self.discovery = discovery
        self._with_usage = with_usage
        self._hive_aware = hive_aware
        self._lock = threading.RLock()
        self._limit_by_type: Dict[str, int] = {}
        self._used_by_type: Dict[str, int] = {}
        self._hive_by_type: Dict[str, int] = {}
        self._total_limit = 0
        self._total_used = 0

    def name(self) -> str:
        return self._name

    def refresh(self) -> None:
        node_inventory = self.discovery.discover()
        by_type: Dict[str, int] = {}
        total = 0
        for _, accels in node_inventory.items():
            for full_name, info in accels.items():
                short = normalize_accelerator_name(full_name)
                by_type[short] = by_type.get(short, 0) + info.count
                total += info.count
        hive = self.discovery.max_hive_by_type() if self._hive_aware else {}
        with self._lock:
            self._limit_by_type = by_type
            self._total_limit = total
            self._hive_by_type = hive

    def refresh_all(self) -> None:
        """Capacity + usage in one operation (RefreshAll)."""
        self.refresh()
        self.set_used(self.discovery.discover_usage())

    def set_used(self, used_by_type: Dict[str, int]) -> None:
        with self._lock:
            self._used_by_type = dict(used_by_type)
            self._total_used = sum(used_by_type.values())

    def create_allocator(self) -> TypeAllocator:
        with self._lock:
            remaining = {
                t: max(limit - self._used_by_type.get(t, 0), 0)
                for t, limit in self._limit_by_type.items()
            }
            return TypeAllocator(remaining, self._hive_by_type)

    def total_limit(self) -> int:
        with self._lock:
            return self._total_limit

    def total_used(self) -> int:
        with self._lock:
            return self._total_used

    def total_available(self) -> int:
        with self._lock:
            return max(self._total_limit - self._total_used, 0)

    def limit_by_type(self, acc_type: str) -> int:
        with self._lock:
            return self._limit_by_type.get(acc_type, 0)

    def used_by_type(self, acc_type: str) -> int:
        with self._lock:
            return self._used_by_type.get(acc_type, 0)

    def available_by_type(self, acc_type: str) -> int:
        with self._lock:
            return max(
                self._limit_by_type.get(acc_type, 0)
                - self._used_by_type.get(acc_type, 0),
                0,
            )

    def get_resource_pools(self) -> Dict[str, ResourcePool]:
        with self._lock:
            pools = {}
            for t, limit in self._limit_by_type.items():
                used = self._used_by_type.get(t, 0)
                pools[t] = ResourcePool(
                    limit=limit, used=used, available=max(limit - used, 0)
                )
            return pools

    def accelerator_types(self):
        with self._lock:
            return list(self._limit_by_type.keys())
