"""System: registry of accelerators, models, service classes, servers,
capacities; per-type allocation aggregation and solution export.

Parity with /root/reference/pkg/core/system.go minus the ``TheSystem``
singleton and its free-function accessors — all lookups go through an
explicit System instance.
"""

from __future__ import annotations

from typing import Dict, Optional

from ..config import (
    AcceleratorCount,
    AcceleratorSpec,
    AllocationSolution,
    ModelAcceleratorPerfData,
    OptimizerSpec,
    ServerSpec,
    ServiceClassSpec,
    SystemSpec,
)
from .accelerator import Accelerator
from .model import Model
from .server import Server
from .serviceclass import ServiceClass


class AllocationByType:
    """Aggregated allocation for one accelerator *type*."""

    def __init__(self, name: str, limit: int) -> None:
        self.name = name
        self.count = 0
        self.limit = limit
        self.cost = 0.0

    def __repr__(self) -> str:
        return f"name={self.name}, count={self.count}, limit={self.limit}, cost={self.cost}"


class System:
    def __init__(self) -> None:
        self.accelerators: Dict[str, Accelerator] = {}
        self.models: Dict[str, Model] = {}
        self.service_classes: Dict[str, ServiceClass] = {}
        self.servers: Dict[str, Server] = {}
        self.capacity: Dict[str, int] = {}
        self.allocation_by_type: Dict[str, AllocationByType] = {}
        self.allocation_solution: Optional[AllocationSolution] = None
        self.optimizer_spec: Optional[OptimizerSpec] = None

    # -- spec loading -------------------------------------------------------
    def set_from_spec(self, spec: SystemSpec) -> OptimizerSpec:
        for acc in spec.accelerators.spec:
            self.add_accelerator(acc)
        for pd in spec.models.perf_data:
            self.add_model_perf_data(pd)
        for sc in spec.service_classes.spec:
            self.add_service_class_from_spec(sc)
        for srv in spec.servers.spec:
            self.add_server(srv)
        for cnt in spec.capacity.count:
            self.set_capacity(cnt)
        self.optimizer_spec = spec.optimizer.spec
        return spec.optimizer.spec

    def add_accelerator(self, spec: AcceleratorSpec) -> None:
        self.accelerators[spec.name] = Accelerator(spec)

    def remove_accelerator(self, name: str) -> None:
        if name not in self.accelerators:
            raise KeyError(f"accelerator {name} not found")
        del self.accelerators[name]

    def add_model_perf_data(self, pd: ModelAcceleratorPerfData) -> Model:
        model = self.models.get(pd.name)
        if model is None:
            model = Model(pd.name)
            self.models[pd.name] = model
        model.add_perf_data(pd)
        return model

    def add_service_class_from_spec(self, spec: ServiceClassSpec) -> None:
        self.service_classes[spec.name] = ServiceClass.from_spec(spec)

    def add_service_class(self, name: str, priority: int) -> None:
        self.service_classes[name] = ServiceClass(name, priority)

    def add_server(self, spec: ServerSpec) -> None:
        self.servers[spec.name] = Server(spec)

    def remove_server(self, name: str) -> None:
        if name not in self.servers:
            raise KeyError(f"server {name} not found")
        del self.servers[name]

    def remove_model(self, name: str) -> None:
        # system.go:570 TestSystem_RemoveModel — removing a missing model
        # raises, matching remove_accelerator/remove_server semantics
        if name not in self.models:
            raise KeyError(f"model {name} not found")
        del self.models[name]

    def remove_service_class(self, name: str) -> None:
        if name not in self.service_classes:
            raise KeyError(f"service class {name} not found")
        del self.service_classes[name]

    def set_capacity(self, cnt: AcceleratorCount) -> None:
        self.capacity[cnt.type] = cnt.count

    def remove_capacity(self, type_name: str) -> None:
        # system.go:1179 TestSystem_RemoveCapacity — no-op when absent
        # (capacity is advisory; limited mode treats a missing type as
        # unconstrained only if no pool was ever declared)
        self.capacity.pop(type_name, None)

    # -- lookups ------------------------------------------------------------
    def accelerator(self, name: str) -> Optional[Accelerator]:
        return self.accelerators.get(name)

    def model(self, name: str) -> Optional[Model]:
        return self.models.get(name)

    def service_class(self, name: str) -> Optional[ServiceClass]:
        return self.service_classes.get(name)

    def server(self, name: str) -> Optional[Server]:
        return self.servers.get(name)

    # -- computation --------------------------------------------------------
    def calculate(self) -> None:
        """Calculate candidate allocations for all servers (the analyze
        phase, hot loop #1 — system.go:258-268)."""
        for g in self.accelerators.values():
            g.calculate()
        for m in self.models.values():
            m.calculate(self.accelerators)
        for v in self.servers.values():
            v.calculate(self, self.accelerators)

    def allocate_by_type(self) -> None:
        """Accumulate per-accelerator-type counts/costs from the solution;
        count += numReplicas * numInstances * multiplicity (system.go:271-300)."""
        self.allocation_by_type = {}
        for server in self.servers.values():
            alloc = server.allocation
            if alloc is None:
                continue
            acc = self.accelerators.get(alloc.accelerator)
            model = self.models.get(server.model_name)
            if acc is None or model is None:
                continue
            t = acc.type
            entry = self.allocation_by_type.get(t)
            if entry is None:
                entry = AllocationByType(t, self.capacity.get(t, 0))
            entry.count += alloc.num_replicas * model.num_instances.get(alloc.accelerator, 0) * acc.multiplicity
            entry.cost += alloc.cost
            self.allocation_by_type[t] = entry

    def generate_solution(self) -> AllocationSolution:
        solution = AllocationSolution(spec={})
        for server_name, server in self.servers.items():
            alloc = server.allocation
            if alloc is None:
                continue
            data = alloc.allocation_data()
            data.load = server.load
            solution.spec[server_name] = data
        self.allocation_solution = solution
        return solution

    def __repr__(self) -> str:
        lines = ["Solution:"]
        total_cost = 0.0
        for name, server in self.servers.items():
            svc = self.service_classes.get(server.service_class_name)
            if server.load is None or svc is None:
                continue
            target = svc.model_target(server.model_name)
            if target is None:
                continue
            alloc = server.allocation
            if alloc is None:
                lines.append(f"s={name}; c={server.service_class_name}; m={server.model_name}; no feasible allocation!")
                continue
            total_cost += alloc.cost
            lines.append(
                f"s={name}; c={server.service_class_name}; m={server.model_name}; "
                f"rate={server.load.arrival_rate}; inTk={server.load.avg_in_tokens}; "
                f"outTk={server.load.avg_out_tokens}; sol={len(server.all_allocations)}, "
                f"sat={server.saturated()}, alloc={alloc}; "
                f"slo-itl={target.itl}, slo-ttft={target.ttft}, slo-tps={target.tps}"
            )
        lines.append("AllocationByType:")
        for a in self.allocation_by_type.values():
            lines.append(repr(a))
        lines.append(f"totalCost={total_cost}")
        return "\n".join(lines)
