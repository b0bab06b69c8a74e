"""Timed optimizer wrapper.

Parity with /root/reference/pkg/solver/optimizer.go — and a deliberate
upgrade: the reference only exposes the solve wall-clock via ``String()``
(optimizer.go:30-34,46) although it is the headline metric; here it is also
recorded in a Prometheus histogram (``wva_solver_duration_seconds``) when
the metrics registry is initialized (see wva_amd.controller.metrics).
"""

from __future__ import annotations

import time
from typing import Optional

from ..config import OptimizerSpec
from ..core import System
from .solver import Solver


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
        self._observe(self.solution_time_msec)

    @staticmethod
    def _observe(msec: float) -> None:
        from ..controller import metrics as ctrl_metrics

        ctrl_metrics.observe_solver_duration(msec / 1000.0)

    def __repr__(self) -> str:
        s = repr(self.solver) + "\n" if self.solver is not None else ""
        return f"{s}Solution time: {self.solution_time_msec:.3f} msec"
