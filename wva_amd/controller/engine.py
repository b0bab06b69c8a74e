"""Global optimization engine adapter.

Parity with /root/reference/internal/optimizer/optimizer.go: run the
manager's optimize pass, export the allocation solution, and map it to
per-VA OptimizedAlloc entries.
"""

from __future__ import annotations

from typing import Dict

from ..api import v1alpha1
from ..core import System
from ..solver import Manager
from .interfaces import ModelAnalyzeResponse
from .logger import log
from .utils import create_optimized_alloc


class OptimizationError(RuntimeError):
    pass


class VariantAutoscalingsEngine:
    def __init__(self, manager: Manager, system: System) -> None:
        self.manager = manager
        self.system = system

    def optimize(
        self,
        va_list: v1alpha1.VariantAutoscalingList,
        analysis: Dict[str, ModelAnalyzeResponse],
    ) -> Dict[str, v1alpha1.OptimizedAlloc]:
        try:
            self.manager.optimize()
        except Exception as e:
            raise OptimizationError(str(e)) from e
        solution = self.system.generate_solution()
        if not solution.spec:
            raise OptimizationError("no feasible allocations found for all variants")
        log.debug("Optimization solution", servers=len(solution.spec))

        out: Dict[str, v1alpha1.OptimizedAlloc] = {}
        for va in va_list.items:
            try:
                out[va.name] = create_optimized_alloc(va.name, va.namespace, solution)
            except KeyError:
                continue
        return out
