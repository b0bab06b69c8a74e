"""Per-variant model analyzer.

Parity with /root/reference/internal/modelanalyzer/{analyzer,utils}.go: the
per-server candidate-allocation calculation adapted into a
ModelAnalyzeResponse with RequiredPrefillQPS = RequiredDecodeQPS =
maxArrvRatePerReplica * 1000 and reason "markovian analysis".

MI355X upgrade: when constructed with ``batched=True`` the whole fleet's
sizing runs through the native batched solver (wva_amd.ops) — one GPU
dispatch instead of per-server scalar sizing; ``analyze_model`` then simply
reads the precomputed allocations.
"""

from __future__ import annotations

from typing import Dict, Optional

from ..api import v1alpha1
from ..core import Allocation, System
from .interfaces import ModelAcceleratorAllocation, ModelAnalyzeResponse
from .utils import full_name


def response_from_allocations(allocations: Dict[str, Allocation]) -> ModelAnalyzeResponse:
    return ModelAnalyzeResponse(
        allocations={
            acc: ModelAcceleratorAllocation(
                allocation=alloc,
                required_prefill_qps=alloc.max_arrv_rate_per_replica * 1000.0,
                required_decode_qps=alloc.max_arrv_rate_per_replica * 1000.0,
                reason="markovian analysis",
            )
            for acc, alloc in allocations.items()
        }
    )


class ModelAnalyzer:
    def __init__(self, system: System, *, batched: bool = False, device: Optional[str] = None) -> None:
        self.system = system
        self._batched_done = False
        self.batched = batched
        self.device = device

    def _ensure_batched(self) -> None:
        if not self._batched_done:
            from ..ops import BatchedAllocationSolver

            BatchedAllocationSolver(device=self.device).calculate(self.system)
            self._batched_done = True

    def analyze_model(self, va: v1alpha1.VariantAutoscaling) -> ModelAnalyzeResponse:
        server_name = full_name(va.name, va.namespace)
        server = self.system.server(server_name)
        if server is None:
            return ModelAnalyzeResponse()
        if self.batched:
            self._ensure_batched()
        else:
            server.calculate(self.system, self.system.accelerators)
        return response_from_allocations(server.all_allocations)
