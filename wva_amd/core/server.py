"""Server: one VariantAutoscaling <=> one inference server (variant).

Parity with /root/reference/pkg/core/server.go, including the
keep-accelerator candidate pinning (server.go:70-82) used by unlimited mode.
"""

from __future__ import annotations

from typing import TYPE_CHECKING, Dict, Optional

from ..config import (
    DEFAULT_SERVICE_CLASS_NAME,
    DEFAULT_SERVICE_CLASS_PRIORITY,
    AllocationData,
    ServerLoadSpec,
    ServerSpec,
)
from .allocation import Allocation, create_allocation, energy_value_term

if TYPE_CHECKING:  # pragma: no cover
    from .accelerator import Accelerator
    from .system import System


class Server:
    def __init__(self, spec: ServerSpec) -> None:
        self.name = spec.name
        self.service_class_name = spec.class_name or DEFAULT_SERVICE_CLASS_NAME
        self.model_name = spec.model
        self.keep_accelerator = spec.keep_accelerator
        self.min_num_replicas = spec.min_num_replicas
        self.max_batch_size = spec.max_batch_size
        # negative = unset -> the global analyzer configuration applies
        self.service_scv = getattr(spec, "service_scv", -1.0)
        self.load: Optional[ServerLoadSpec] = spec.current_alloc.load
        self.all_allocations: Dict[str, Allocation] = {}
        self.allocation: Optional[Allocation] = None
        self.cur_allocation: Optional[Allocation] = Allocation.from_data(spec.current_alloc)
        self.spec = spec

    def calculate(self, system: "System", accelerators: Dict[str, "Accelerator"]) -> None:
        """Enumerate candidate allocations; value = transition penalty from
        the current allocation (server.go:55-67), plus the optional
        energy term of the cost+energy objective."""
        candidates = self.get_candidate_accelerators(accelerators)
        self.all_allocations = {}
        for g in candidates.values():
            alloc = create_allocation(system, self.name, g.name)
            if alloc is not None:
                if self.cur_allocation is not None:
                    alloc.set_value(self.cur_allocation.transition_penalty(alloc))
                alloc.set_value(alloc.value + energy_value_term(system, self, alloc))
                self.all_allocations[g.name] = alloc

    def get_candidate_accelerators(
        self, accelerators: Dict[str, "Accelerator"]
    ) -> Dict[str, "Accelerator"]:
        if self.keep_accelerator and self.cur_allocation is not None and self.cur_allocation.accelerator:
            cur = self.cur_allocation.accelerator
            acc = accelerators.get(cur)
            return {cur: acc} if acc is not None else {}
        return accelerators

    def priority(self, system: "System") -> int:
        svc = system.service_class(self.service_class_name)
        return svc.priority if svc is not None else DEFAULT_SERVICE_CLASS_PRIORITY

    def set_load(self, load: ServerLoadSpec) -> None:
        self.load = load

    def set_allocation(self, alloc: Allocation) -> None:
        self.allocation = alloc
        self.update_desired_alloc()

    def remove_allocation(self) -> None:
        self.allocation = None

    def set_cur_allocation(self, alloc: Optional[Allocation]) -> None:
        self.cur_allocation = alloc

    def saturated(self) -> bool:
        return (
            self.allocation is not None
            and self.load is not None
            and self.allocation.saturated(self.load.arrival_rate)
        )

    def update_desired_alloc(self) -> None:
        if self.allocation is not None:
            self.spec.desired_alloc = self.allocation.allocation_data()
            self.spec.desired_alloc.load = self.load
        else:
            self.spec.desired_alloc = AllocationData()

    def apply_desired_alloc(self) -> None:
        self.spec.current_alloc = self.spec.desired_alloc
        self.cur_allocation = Allocation.from_data(self.spec.current_alloc)
        self.load = self.spec.current_alloc.load

    def __repr__(self) -> str:
        return (
            f"Server: name={self.name}; class={self.service_class_name}; "
            f"model={self.model_name}; load={self.load}; allocation={self.allocation}"
        )
