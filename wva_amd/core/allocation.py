"""Allocation — the sizing kernel.

For a (server, accelerator) pair: derive the batch size N, build a queue
analyzer, find the max per-replica rate meeting the SLO targets, compute
replicas = ceil(totalRate / rate*), cost, and predicted ITL/TTFT/rho.

Parity with /root/reference/pkg/core/allocation.go:27-300, preserving the
de-facto contract quirks:
- N falls back to max(perf.maxBatchSize * atTokens / K, 1) when the server
  has no maxBatchSize override; the controller path never sets atTokens so
  the VA's maxBatchSize is effectively mandatory (allocation.go:77-87);
- maxQueue = 10*N; total rate is arrivalRate/60 (req/min -> req/s) unless a
  TPS target overrides it as TPS/K (allocation.go:134-139);
- maxArrvRatePerReplica is stored in req/ms (rate*/1000, allocation.go:160);
- transition penalty = cost delta plus 0.1*(costA+costB) when the
  accelerator type changes (allocation.go:291-300).
"""

from __future__ import annotations

import math
import os
from typing import TYPE_CHECKING, Optional, Tuple

from ..analyzer import (
    AnalyzerError,
    Configuration,
    DecodeParms,
    PrefillParms,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    TargetPerf,
)
from ..config import (
    ACCEL_PENALTY_FACTOR,
    MAX_QUEUE_TO_BATCH_RATIO,
    AllocationData,
)

if TYPE_CHECKING:  # pragma: no cover
    from .system import System


class Allocation:
    """Details of an accelerator allocated to a server."""

    __slots__ = (
        "accelerator", "num_replicas", "batch_size", "cost", "value",
        "itl", "ttft", "rho", "max_arrv_rate_per_replica",
    )

    def __init__(
        self,
        accelerator: str = "",
        num_replicas: int = 0,
        batch_size: int = 0,
        cost: float = 0.0,
        itl: float = 0.0,
        ttft: float = 0.0,
        rho: float = 0.0,
        max_arrv_rate_per_replica: float = 0.0,  # req/ms
    ) -> None:
        self.accelerator = accelerator
        self.num_replicas = num_replicas
        self.batch_size = batch_size
        self.cost = cost
        self.value = 0.0
        self.itl = itl
        self.ttft = ttft
        self.rho = rho
        self.max_arrv_rate_per_replica = max_arrv_rate_per_replica

    # -- accessors mirroring the reference API ------------------------------
    def set_value(self, value: float) -> None:
        self.value = value

    def max_rpm(self) -> float:
        return self.max_arrv_rate_per_replica * 1000.0 * 60.0

    def saturated(self, total_rate_per_min: float) -> bool:
        return total_rate_per_min > self.num_replicas * self.max_rpm()

    def transition_penalty(self, b: "Allocation") -> float:
        """Penalty for transitioning from this allocation to ``b``."""
        if self.accelerator == b.accelerator:
            if self.num_replicas == b.num_replicas:
                return 0.0
            return b.cost - self.cost
        return ACCEL_PENALTY_FACTOR * (self.cost + b.cost) + (b.cost - self.cost)

    def clone(self) -> "Allocation":
        a = Allocation(
            accelerator=self.accelerator,
            num_replicas=self.num_replicas,
            batch_size=self.batch_size,
            cost=self.cost,
            itl=self.itl,
            ttft=self.ttft,
            rho=self.rho,
            max_arrv_rate_per_replica=self.max_arrv_rate_per_replica,
        )
        a.value = self.value
        return a

    def allocation_data(self) -> AllocationData:
        return AllocationData(
            accelerator=self.accelerator,
            num_replicas=self.num_replicas,
            max_batch=self.batch_size,
            cost=self.cost,
            itl_average=self.itl,
            ttft_average=self.ttft,
        )

    @classmethod
    def from_data(cls, data: AllocationData) -> "Allocation":
        return cls(
            accelerator=data.accelerator,
            num_replicas=data.num_replicas,
            batch_size=data.max_batch,
            cost=data.cost,
            itl=data.itl_average,
            ttft=data.ttft_average,
        )

    def __repr__(self) -> str:
        return (
            f"{{acc={self.accelerator}; numRep={self.num_replicas}; maxBatch={self.batch_size}; "
            f"cost={self.cost}, val={self.value}, itl={self.itl}, ttft={self.ttft}, "
            f"rho={self.rho}, maxRPM={self.max_rpm()}}}"
        )


def sizing_headroom() -> float:
    """Opt-in provisioning headroom (WVA_SIZING_HEADROOM, e.g. "0.25"):
    the fleet is sized for measured load x (1 + h), absorbing ramp
    transients between reconcile cycles (the endurance soaks' only
    SLO misses were stages served at the PREVIOUS cycle's size).  The
    reference has no equivalent; 0 (default) keeps exact parity.
    Measured load reported in status stays untouched — headroom applies
    to sizing only."""
    raw = os.environ.get("WVA_SIZING_HEADROOM", "").strip()
    if not raw:
        return 0.0
    try:
        h = float(raw)
    except ValueError:
        return 0.0
    return max(h, 0.0)


def create_allocation(system: "System", server_name: str, acc_name: str) -> Optional[Allocation]:
    """Create an allocation of accelerator ``acc_name`` to ``server_name``;
    ``None`` if infeasible.  Takes the system explicitly (no singleton)."""
    acc = system.accelerator(acc_name)
    if acc is None:
        return None
    server = system.server(server_name)
    if server is None:
        return None
    load = server.load
    if load is None or load.arrival_rate < 0 or load.avg_in_tokens < 0 or load.avg_out_tokens < 0:
        return None
    model = system.model(server.model_name)
    if model is None:
        return None
    perf = model.get_perf_data(acc_name)
    if perf is None:
        return None
    svc = system.service_class(server.service_class_name)
    if svc is None:
        return None
    target = svc.model_target(server.model_name)
    if target is None:
        return None

    # zero traffic case
    if load.arrival_rate == 0 or load.avg_out_tokens == 0:
        return _zero_load_allocation(server, model, acc, perf)

    K = int(load.avg_out_tokens)
    if server.max_batch_size > 0:
        N = server.max_batch_size
    else:
        N = max(perf.max_batch_size * perf.at_tokens // K, 1)
    max_queue = N * MAX_QUEUE_TO_BATCH_RATIO

    config = Configuration(
        max_batch_size=N,
        max_queue_size=max_queue,
        service_parms=ServiceParms(
            prefill=PrefillParms(gamma=perf.prefill_parms.gamma, delta=perf.prefill_parms.delta),
            decode=DecodeParms(alpha=perf.decode_parms.alpha, beta=perf.decode_parms.beta),
        ),
    )
    request_size = RequestSize(avg_input_tokens=int(load.avg_in_tokens), avg_output_tokens=K)
    try:
        from ..analyzer.mg1 import configured_scv

        # per-server cs^2 (mg1 auto mode, measured token histogram)
        # overrides the fleet-wide setting; negative = unset
        server_scv = getattr(server, "service_scv", -1.0)
        qa = QueueAnalyzer(
            config,
            request_size,
            scv=server_scv if server_scv >= 0 else configured_scv(),
        )
    except AnalyzerError:
        return None

    target_perf = TargetPerf(
        target_ttft=target.ttft, target_itl=target.itl, target_tps=target.tps
    )
    try:
        _, metrics, _ = qa.size(target_perf)
    except AnalyzerError:
        return None
    rate_star = metrics.throughput  # req/s

    if target.tps == 0:
        total_rate = load.arrival_rate / 60.0  # req/min -> req/s
    else:
        total_rate = target.tps / float(K)
    total_rate *= 1.0 + sizing_headroom()
    num_replicas = int(math.ceil(total_rate / rate_star))
    num_replicas = max(num_replicas, server.min_num_replicas)

    total_num_instances = model.get_num_instances(acc_name) * num_replicas
    cost = acc.cost * total_num_instances

    # analyze the queue of a single replica at its share of the load
    rate = total_rate / num_replicas
    try:
        metrics = qa.analyze(rate)
    except AnalyzerError:
        return None

    alloc = Allocation(
        accelerator=acc_name,
        num_replicas=num_replicas,
        batch_size=N,
        cost=cost,
        itl=metrics.avg_token_time,
        ttft=metrics.avg_wait_time + metrics.avg_prefill_time,
        rho=metrics.rho,
        max_arrv_rate_per_replica=rate_star / 1000.0,
    )
    alloc.set_value(alloc.cost)
    return alloc


def energy_value_term(system: "System", server, alloc: Allocation) -> float:
    """Energy surcharge for the ``cost+energy`` objective (an MI355X-native
    extension: the reference computes the accelerator power curve but never
    uses it — accelerator.go:35-41 / SURVEY.md §2a).

    Predicted draw = power(rho) per card x cards per replica x replicas,
    priced at energyCostPerKWh (cents/kWh) -> cents/hr, the same unit as
    the accelerator cost, so it composes with cost- and penalty-based
    values.  Returns 0 unless the objective is enabled.
    """
    spec = system.optimizer_spec
    if spec is None or spec.objective != "cost+energy" or spec.energy_cost_per_kwh <= 0:
        return 0.0
    if alloc.num_replicas == 0 or not alloc.accelerator:
        return 0.0
    acc = system.accelerator(alloc.accelerator)
    model = system.model(server.model_name)
    if acc is None or model is None:
        return 0.0
    cards = model.get_num_instances(alloc.accelerator) * acc.multiplicity * alloc.num_replicas
    watts = acc.power(alloc.rho) * cards
    return watts / 1000.0 * spec.energy_cost_per_kwh  # cents per hour


def scale_allocation(
    system: "System", alloc: Allocation, server_name: str
) -> Tuple[Optional[Allocation], int]:
    """Re-size this allocation for the server's current load; returns the
    new allocation and the replica increment (allocation.go:166-190)."""
    server = system.server(server_name)
    if server is None or server.load is None:
        return None, 0
    if system.accelerator(alloc.accelerator) is None:
        return None, 0
    new_alloc = create_allocation(system, server_name, alloc.accelerator)
    if new_alloc is None:
        return None, 0
    return new_alloc, new_alloc.num_replicas - alloc.num_replicas


def reallocate(system: "System", server_name: str) -> Tuple[Optional[Allocation], str]:
    """Pick the min-value feasible allocation across all accelerators
    (allocation.go:192-207)."""
    min_val = 0.0
    min_alloc: Optional[Allocation] = None
    for acc_name in system.accelerators:
        alloc = create_allocation(system, server_name, acc_name)
        # quirk preserved from allocation.go:197: a zero min value keeps
        # accepting replacements
        if alloc is not None and (min_alloc is None or min_val == 0 or alloc.value < min_val):
            min_val = alloc.value
            min_alloc = alloc
    if min_alloc is None:
        return None, ""
    return min_alloc, min_alloc.accelerator


def _zero_load_allocation(server, model, acc, perf) -> Allocation:
    """Allocation in case of zero load (allocation.go:259-288)."""
    num_replicas = server.min_num_replicas
    if num_replicas == 0:
        alloc = Allocation()
        alloc.set_value(0.0)
        return alloc

    max_batch_size = perf.max_batch_size
    if server.max_batch_size > 0:
        max_batch_size = server.max_batch_size
    total_num_instances = model.get_num_instances(acc.name) * num_replicas
    cost = acc.cost * total_num_instances

    decode_time = perf.decode_parms.alpha + perf.decode_parms.beta
    max_decode_time = perf.decode_parms.alpha + perf.decode_parms.beta * max_batch_size
    prefill_time = perf.prefill_parms.gamma + perf.prefill_parms.delta
    max_serv_time = prefill_time + max_decode_time
    max_arrv_rate = max_batch_size / max_serv_time if max_serv_time > 0 else 0.0

    alloc = Allocation(
        accelerator=acc.name,
        num_replicas=num_replicas,
        batch_size=max_batch_size,
        cost=cost,
        itl=decode_time,
        ttft=prefill_time,
        rho=0.0,
        max_arrv_rate_per_replica=max_arrv_rate,
    )
    alloc.set_value(alloc.cost)
    return alloc


class AllocationDiff:
    """Orchestration difference between two allocations."""

    def __init__(self, a: Optional[Allocation], b: Optional[Allocation]) -> None:
        self.old_accelerator = a.accelerator if a else "none"
        self.new_accelerator = b.accelerator if b else "none"
        self.old_num_replicas = a.num_replicas if a else 0
        self.new_num_replicas = b.num_replicas if b else 0
        self.cost_diff = (b.cost if b else 0.0) - (a.cost if a else 0.0)

    @classmethod
    def create(cls, a: Optional[Allocation], b: Optional[Allocation]) -> Optional["AllocationDiff"]:
        if a is None and b is None:
            return None
        return cls(a, b)

    def __repr__(self) -> str:
        return (
            f"{{ {self.old_accelerator} -> {self.new_accelerator}, "
            f"{self.old_num_replicas} -> {self.new_num_replicas}, {self.cost_diff} }}"
        )
