"""Inferno solver: unlimited + greedy capacity-constrained allocation.

Parity: reference pkg/solver/{solver,greedy,optimizer}.go —
  * SolveUnlimited: independent min-value allocation per server
  * SolveGreedy: per-server candidate lists sorted by value ascending;
    server entries ordered by (priority asc, delta-regret desc, current
    value desc); allocation walks entries, falling to the next candidate
    when the accelerator-type pool can't cover it; unallocated servers get
    best-effort treatment per saturation policy (PriorityExhaustive /
    PriorityRoundRobin / RoundRobin / None), with cost/value rescaled by
    the granted-replica fraction.
"""
from __future__ import annotations

import bisect
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .system import Allocation, AllocationDiff, Server, System
from .types import (
    OptimizerSpec,
    POLICY_NONE,
    POLICY_PRIORITY_EXHAUSTIVE,
    POLICY_PRIORITY_ROUND_ROBIN,
    POLICY_ROUND_ROBIN,
)

_INF = float("inf")


@dataclass
class _ServerEntry:
    server_name: str
    priority: int
    allocations: List[Allocation] = field(default_factory=list)
    cur_index: int = 0
    delta: float = 0.0

    def current(self) -> Allocation:
        return self.allocations[self.cur_index]


def _order_key(e: _ServerEntry):
    # priority asc, then delta desc, then current value desc
    return (e.priority, -e.delta, -e.current().value)


class Solver:
    def __init__(self, optimizer_spec: Optional[OptimizerSpec] = None):
        self.spec = optimizer_spec or OptimizerSpec()
        self.current_allocation: Dict[str, Allocation] = {}
        self.diff_allocation: Dict[str, AllocationDiff] = {}
        self.solve_time_ms: float = 0.0

    def solve(self, system: System) -> None:
        t0 = time.perf_counter()
        self.current_allocation = {
            name: server.allocation
            for name, server in system.servers.items()
            if server.allocation is not None
        }
        if self.spec.unlimited:
            self._solve_unlimited(system)
        else:
            self._solve_greedy(system)

        self.diff_allocation = {}
        for name, server in system.servers.items():
            cur = self.current_allocation.get(name)
            desired = server.allocation
            if cur is None and desired is None:
                continue
            diff = AllocationDiff(
                accelerator_from=cur.accelerator if cur else "",
                accelerator_to=desired.accelerator if desired else "",
                replicas_from=cur.num_replicas if cur else 0,
                replicas_to=desired.num_replicas if desired else 0,
            )
            if diff.is_change:
                self.diff_allocation[name] = diff
        self.solve_time_ms = (time.perf_counter() - t0) * 1000.0

    # --- unlimited (solver.go:63-79) ---

    def _solve_unlimited(self, system: System) -> None:
        for server in system.servers.values():
            server.remove_allocation()
            best = None
            for alloc in server.all_allocations.values():
                if best is None or alloc.value < best.value:
                    best = alloc
            if best is not None:
                server.set_allocation(best)

    # --- greedy (greedy.go:35-104) ---

    def _solve_greedy(self, system: System) -> None:
        available = dict(system.capacity)
        entries: List[_ServerEntry] = []
        for name, server in system.servers.items():
            server.remove_allocation()
            allocs = sorted(
                server.all_allocations.values(), key=lambda a: a.value
            )
            if not allocs:
                continue
            e = _ServerEntry(
                server_name=name, priority=server.priority, allocations=allocs
            )
            e.delta = (
                allocs[1].value - allocs[0].value if len(allocs) > 1 else _INF
            )
            entries.append(e)
        entries.sort(key=_order_key)

        if self.spec.delayed_best_effort:
            unallocated = self._allocate(system, entries, available)
            self._best_effort(system, unallocated, available)
        else:
            for group in _priority_groups(entries):
                unallocated = self._allocate(system, group, available)
                self._best_effort(system, unallocated, available)

    def _allocate(
        self,
        system: System,
        entries: List[_ServerEntry],
        available: Dict[str, int],
    ) -> List[_ServerEntry]:
        unallocated: List[_ServerEntry] = []
        entries = list(entries)
        while entries:
            top = entries.pop(0)
            if not top.allocations:
                continue
            server = system.servers.get(top.server_name)
            if server is None:
                continue
            alloc = top.allocations[top.cur_index]
            acc = system.accelerators.get(alloc.accelerator)
            if acc is None:
                continue
            units = system.units_per_replica(server.spec.model, alloc.accelerator)
            count = alloc.num_replicas * units
            if available.get(acc.type, 0) >= count:
                available[acc.type] = available.get(acc.type, 0) - count
                server.set_allocation(alloc)
            else:
                top.cur_index += 1
                if top.cur_index + 1 < len(top.allocations):
                    top.delta = (
                        top.allocations[top.cur_index + 1].value
                        - top.allocations[top.cur_index].value
                    )
                elif top.cur_index == len(top.allocations):
                    unallocated.append(top)
                    continue
                else:
                    top.delta = _INF
                keys = [_order_key(e) for e in entries]
                idx = bisect.bisect_left(keys, _order_key(top))
                entries.insert(idx, top)
        return unallocated

    # --- best effort (greedy.go bestEffort) ---

    def _best_effort(
        self,
        system: System,
        unallocated: List[_ServerEntry],
        available: Dict[str, int],
    ) -> None:
        policy = self.spec.saturation_policy
        if policy == POLICY_PRIORITY_EXHAUSTIVE:
            self._allocate_maximally(system, unallocated, available)
        elif policy == POLICY_PRIORITY_ROUND_ROBIN:
            for group in _priority_groups(unallocated):
                self._allocate_equally(system, group, available)
        elif policy == POLICY_ROUND_ROBIN:
            self._allocate_equally(system, unallocated, available)
        elif policy == POLICY_NONE:
            return

    def _allocate_maximally(
        self,
        system: System,
        entries: List[_ServerEntry],
        available: Dict[str, int],
    ) -> None:
        for entry in entries:
            server = system.servers.get(entry.server_name)
            if server is None:
                continue
            for alloc in entry.allocations:
                acc = system.accelerators.get(alloc.accelerator)
                if acc is None:
                    continue
                units = system.units_per_replica(
                    server.spec.model, alloc.accelerator
                )
                if units <= 0:
                    continue
                max_replicas = min(
                    available.get(acc.type, 0) // units, alloc.num_replicas
                )
                if max_replicas > 0:
                    factor = max_replicas / alloc.num_replicas
                    alloc.cost *= factor
                    alloc.value *= factor
                    alloc.num_replicas = max_replicas
                    server.set_allocation(alloc)
                    available[acc.type] -= max_replicas * units
                    break

    def _allocate_equally(
        self,
        system: System,
        entries: List[_ServerEntry],
        available: Dict[str, int],
    ) -> None:
        @dataclass
        class Ticket:
            entry: _ServerEntry
            server: Server
            active: bool = False
            acc_type: str = ""
            units: int = 0
            num_replicas: int = 0
            final_alloc: Optional[Allocation] = None

        tickets: Dict[str, Ticket] = {}
        for entry in entries:
            server = system.servers.get(entry.server_name)
            if server is None:
                continue
            tickets[entry.server_name] = Ticket(entry=entry, server=server)

        allocated: Dict[str, Ticket] = {}
        while tickets:
            for entry in entries:
                ticket = tickets.get(entry.server_name)
                if ticket is None:
                    continue
                if not ticket.active:
                    for alloc in entry.allocations:
                        acc = system.accelerators.get(alloc.accelerator)
                        if acc is None:
                            continue
                        units = system.units_per_replica(
                            ticket.server.spec.model, alloc.accelerator
                        )
                        if units > 0 and available.get(acc.type, 0) >= units:
                            ticket.active = True
                            ticket.acc_type = acc.type
                            ticket.units = units
                            ticket.final_alloc = alloc
                            break
                    if not ticket.active:
                        del tickets[entry.server_name]
                        continue
                replicas_available = available.get(ticket.acc_type, 0) // ticket.units
                allocatable = min(
                    replicas_available,
                    ticket.final_alloc.num_replicas - ticket.num_replicas,
                )
                if allocatable > 0:
                    ticket.num_replicas += 1
                    available[ticket.acc_type] -= ticket.units
                    allocated[entry.server_name] = ticket
                else:
                    del tickets[entry.server_name]

        for ticket in allocated.values():
            alloc = ticket.final_alloc
            factor = ticket.num_replicas / alloc.num_replicas
            alloc.cost *= factor
            alloc.value *= factor
            alloc.num_replicas = ticket.num_replicas
            ticket.server.set_allocation(alloc)


def _priority_groups(entries: List[_ServerEntry]) -> List[List[_ServerEntry]]:
    groups: List[List[_ServerEntry]] = []
    i = 0
    while i < len(entries):
        group = [entries[i]]
        prio = entries[i].priority
        i += 1
        while i < len(entries) and entries[i].priority == prio:
            group.append(entries[i])
            i += 1
        groups.append(group)
    return groups
