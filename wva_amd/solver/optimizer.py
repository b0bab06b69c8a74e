"""Timed optimizer wrapper.

Parity with /root/reference/pkg/solver/optimizer.go — and a deliberate
upgrade: the reference only exposes the solve wall-clock via ``String()``
(optimizer.go:30-34,46) although it is the headline metric; here every
solve also notifies registered observers (the controller's metrics layer
registers its Prometheus histogram via :func:`register_solve_observer`,
keeping the dependency direction strictly downward).
"""

from __future__ import annotations

import time
from typing import Callable, List, Optional

from ..config import OptimizerSpec
from ..core import System
from .solver import Solver

# solve-duration observers (seconds); upper layers register callbacks so
# this layer never imports them
_solve_observers: List[Callable[[float], None]] = []


def register_solve_observer(fn: Callable[[float], None]) -> None:
    if fn not in _solve_observers:
        _solve_observers.append(fn)


def unregister_solve_observer(fn: Callable[[float], None]) -> None:
    if fn in _solve_observers:
        _solve_observers.remove(fn)


class Optimizer:
    def __init__(self, spec: OptimizerSpec) -> None:
        self.spec = spec
        self.solver: Optional[Solver] = None
        self.solution_time_msec: float = 0.0

    def optimize(self, system: System) -> None:
        if self.spec is None:
            raise ValueError("missing optimizer spec")
        self.solver = Solver(self.spec)
        start = time.perf_counter()
        self.solver.solve(system)
        self.solution_time_msec = (time.perf_counter() - start) * 1000.0
        for observer in _solve_observers:
            observer(self.solution_time_msec / 1000.0)

    def __repr__(self) -> str:
        s = repr(self.solver) + "\n" if self.solver is not None else ""
        return f"{s}Solution time: {self.solution_time_msec:.3f} msec"
