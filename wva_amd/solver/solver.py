"""Solver frontend: snapshot current allocations, dispatch unlimited vs
greedy, compute per-server allocation diffs.

Parity with /root/reference/pkg/solver/solver.go (the System is passed
explicitly instead of read from a singleton).
"""

from __future__ import annotations

import math
from typing import Dict, Optional

from ..config import OptimizerSpec
from ..core import Allocation, AllocationDiff, System
from .greedy import solve_greedy


class Solver:
    def __init__(self, optimizer_spec: OptimizerSpec) -> None:
        self.optimizer_spec = optimizer_spec
        self.current_allocation: Dict[str, Allocation] = {}
        self.diff_allocation: Dict[str, AllocationDiff] = {}

    def solve(self, system: System) -> None:
        # snapshot current allocations
        self.current_allocation = {
            name: server.cur_allocation
            for name, server in system.servers.items()
            if server.cur_allocation is not None
        }

        if self.optimizer_spec.unlimited:
            self.solve_unlimited(system)
        else:
            solve_greedy(system, self.optimizer_spec)

        self.diff_allocation = {}
        for name, server in system.servers.items():
            diff = AllocationDiff.create(self.current_allocation.get(name), server.allocation)
            if diff is not None:
                self.diff_allocation[name] = diff

    def solve_unlimited(self, system: System) -> None:
        """Separable objective: per-server argmin of allocation value
        (solver.go:63-79)."""
        for server in system.servers.values():
            server.remove_allocation()
            min_val = math.inf
            min_alloc: Optional[Allocation] = None
            for alloc in server.all_allocations.values():
                if alloc.value < min_val:
                    min_val = alloc.value
                    min_alloc = alloc
            if min_alloc is not None:
                server.set_allocation(min_alloc)

    def __repr__(self) -> str:
        lines = ["Solver:"]
        for name, diff in self.diff_allocation.items():
            lines.append(f"sName={name}, allocDiff={diff}")
        return "\n".join(lines)
