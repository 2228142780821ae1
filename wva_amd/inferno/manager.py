"""Inferno manager: optimize = generate allocations + solve.

Parity: reference pkg/manager/manager.go:13-27.
"""
from __future__ import annotations

from typing import Dict, Optional

from .solver import Solver
from .system import AllocationDiff, System
from .types import OptimizerSpec


class Manager:
    def __init__(self, system: System, optimizer_spec: Optional[OptimizerSpec] = None):
        self.system = system
        self.solver = Solver(optimizer_spec)

    def optimize(self) -> Dict[str, AllocationDiff]:
        """Generate all candidate allocations, solve, return the diff from
        the previous allocation per server."""
        self.system.generate_all_allocations()
        self.solver.solve(self.system)
        return self.solver.diff_allocation
