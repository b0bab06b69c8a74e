"""Port of the reference's adversarial greedy-solver matrix
(/root/reference/pkg/solver/greedy_test.go, 1,696 LoC — VERDICT r01 #5).

Fixture mirrors setupTestSystemForGreedy (greedy_test.go:13-207): two
accelerator types (their A100/H100 become the MI300X/MI355X pair, same
costs 1.0/2.0 and capacities 4/2), two models with per-acc profiles
(accCount 1 vs 2), three priority classes, three base servers.  Test
names trace to the Go functions they port.
"""

import pytest

from wva_amd.config import (
    AcceleratorCount,
    AcceleratorData,
    AcceleratorSpec,
    AllocationData,
    CapacityData,
    DecodeParmsSpec,
    ModelAcceleratorPerfData,
    ModelData,
    ModelTarget,
    OptimizerData,
    OptimizerSpec,
    PowerSpec,
    PrefillParmsSpec,
    ServerData,
    ServerLoadSpec,
    ServerSpec,
    ServiceClassData,
    ServiceClassSpec,
    SystemSpec,
)
from wva_amd.core import System
from wva_amd.solver import Solver
from wva_amd.solver.greedy import (
    ServerEntry,
    _allocate,
    _allocate_equally,
    _allocate_maximally,
    _best_effort,
    make_priority_groups,
    solve_greedy,
)

MI300X_T = "AMD-MI300X-192GB"  # the fixture's 'A100' analog: cost 1.0, cap 4
MI355X_T = "AMD-MI355X-288GB"  # the fixture's 'H100' analog: cost 2.0, cap 2


def fixture_server(
    name,
    model="llama-8b",
    cls="high-priority",
    rate=30.0,
    in_tok=100,
    out_tok=200,
    min_replicas=1,
    max_batch=512,
):
    return ServerSpec(
        name=name,
        class_name=cls,
        model=model,
        min_num_replicas=min_replicas,
        max_batch_size=max_batch,
        current_alloc=AllocationData(
            load=ServerLoadSpec(
                arrival_rate=rate, avg_in_tokens=in_tok, avg_out_tokens=out_tok
            )
        ),
    )


BASE_SERVERS = [
    fixture_server("server1", rate=30, in_tok=100, out_tok=200, max_batch=512),
    fixture_server("server2", model="llama-70b", cls="medium-priority",
                   rate=20, in_tok=150, out_tok=300, max_batch=256),
    fixture_server("server3", cls="low-priority",
                   rate=10, in_tok=80, out_tok=150, max_batch=128),
]


def greedy_system(
    servers=None,
    capacity=((MI300X_T, 4), (MI355X_T, 2)),
    policy="None",
    delayed=False,
    calculate=True,
):
    """setupTestSystemForGreedy analog (greedy_test.go:13-207)."""
    spec = SystemSpec(
        accelerators=AcceleratorData(
            spec=[
                AcceleratorSpec(
                    name="MI300X", type=MI300X_T, multiplicity=1, mem_size=192,
                    power=PowerSpec(idle=50, full=350, mid_power=150, mid_util=0.4),
                    cost=1.0,
                ),
                AcceleratorSpec(
                    name="MI355X", type=MI355X_T, multiplicity=1, mem_size=288,
                    power=PowerSpec(idle=60, full=450, mid_power=200, mid_util=0.5),
                    cost=2.0,
                ),
            ]
        ),
        models=ModelData(
            perf_data=[
                ModelAcceleratorPerfData(
                    name="llama-8b", acc="MI300X", acc_count=1, max_batch_size=16,
                    at_tokens=100,
                    decode_parms=DecodeParmsSpec(alpha=10.0, beta=2.0),
                    prefill_parms=PrefillParmsSpec(gamma=5.0, delta=0.1),
                ),
                ModelAcceleratorPerfData(
                    name="llama-8b", acc="MI355X", acc_count=1, max_batch_size=32,
                    at_tokens=100,
                    decode_parms=DecodeParmsSpec(alpha=8.0, beta=1.5),
                    prefill_parms=PrefillParmsSpec(gamma=3.0, delta=0.08),
                ),
                ModelAcceleratorPerfData(
                    name="llama-70b", acc="MI300X", acc_count=2, max_batch_size=8,
                    at_tokens=150,
                    decode_parms=DecodeParmsSpec(alpha=15.0, beta=3.0),
                    prefill_parms=PrefillParmsSpec(gamma=8.0, delta=0.15),
                ),
                ModelAcceleratorPerfData(
                    name="llama-70b", acc="MI355X", acc_count=1, max_batch_size=16,
                    at_tokens=150,
                    decode_parms=DecodeParmsSpec(alpha=12.0, beta=2.5),
                    prefill_parms=PrefillParmsSpec(gamma=6.0, delta=0.12),
                ),
            ]
        ),
        service_classes=ServiceClassData(
            spec=[
                ServiceClassSpec(
                    name="high-priority", priority=1,
                    model_targets=[
                        ModelTarget(model="llama-8b", slo_itl=400, slo_ttft=20, slo_tps=15),
                        ModelTarget(model="llama-70b", slo_itl=500, slo_ttft=25, slo_tps=12),
                    ],
                ),
                ServiceClassSpec(
                    name="medium-priority", priority=2,
                    model_targets=[
                        ModelTarget(model="llama-8b", slo_itl=450, slo_ttft=22, slo_tps=13),
                        ModelTarget(model="llama-70b", slo_itl=550, slo_ttft=28, slo_tps=10),
                    ],
                ),
                ServiceClassSpec(
                    name="low-priority", priority=3,
                    model_targets=[
                        ModelTarget(model="llama-8b", slo_itl=500, slo_ttft=25, slo_tps=10),
                    ],
                ),
            ]
        ),
        servers=ServerData(spec=servers if servers is not None else list(BASE_SERVERS)),
        optimizer=OptimizerData(
            spec=OptimizerSpec(
                unlimited=False, saturation_policy=policy, delayed_best_effort=delayed
            )
        ),
        capacity=CapacityData(
            count=[AcceleratorCount(type=t, count=c) for t, c in capacity]
        ),
    )
    system = System()
    opt = system.set_from_spec(spec)
    if calculate:
        system.calculate()
    return system, opt


def allocated_servers(system):
    return [s for s in system.servers.values() if s.allocation is not None]


def consumed_units(system):
    """GPU units consumed per type by the chosen allocations."""
    used = {}
    for s in system.servers.values():
        a = s.allocation
        if a is None:
            continue
        acc = system.accelerator(a.accelerator)
        m = system.model(s.model_name)
        units = a.num_replicas * m.get_num_instances(a.accelerator) * acc.multiplicity
        used[acc.type] = used.get(acc.type, 0) + units
    return used


def first_entry(system, name, replicas=None, value=None, count=1):
    """serverEntry for `name` built from its calculated candidates (the
    Go tests' pattern of taking one allocation and pinning replicas)."""
    server = system.servers[name]
    allocs = []
    for a in server.all_allocations.values():
        if replicas is not None:
            a.num_replicas = replicas
        if value is not None:
            a.set_value(value + 10.0 * len(allocs))
        allocs.append(a)
        if len(allocs) >= count:
            break
    return ServerEntry(name, server.priority(system), allocs)


class TestServerEntryString:
    def test_repr_carries_state(self):
        # greedy_test.go:210 TestServerEntry_String
        e = ServerEntry("s", 1, [])
        e.delta = 2.5
        r = repr(e)
        assert "s" in r and "prio=1" in r and "delta=2.5" in r


class TestSolveGreedyScenarios:
    def test_no_servers(self):
        # greedy_test.go:237 TestSolver_SolveGreedy_NoServers
        system = System()
        solve_greedy(system, OptimizerSpec(unlimited=False, saturation_policy="None"))

    def test_basic_allocation(self):
        # greedy_test.go:252 TestSolver_SolveGreedy_BasicAllocation
        servers = list(BASE_SERVERS)
        servers[0] = fixture_server("server1", rate=30, max_batch=16)
        system, opt = greedy_system(servers=servers)
        solve_greedy(system, opt)
        assert len(system.servers["server1"].all_allocations) > 0
        assert system.servers["server1"].allocation is not None

    def test_priority_exhaustive(self):
        # greedy_test.go:410 TestSolver_SolveGreedy_PriorityExhaustive
        servers = [
            fixture_server("server1", rate=10, max_batch=16),
            fixture_server("server2", rate=10, max_batch=16),
        ]
        system, opt = greedy_system(
            servers=servers, policy="PriorityExhaustive", delayed=True
        )
        solve_greedy(system, opt)
        assert len(allocated_servers(system)) >= 1

    def test_priority_round_robin(self):
        # greedy_test.go:485 TestSolver_SolveGreedy_PriorityRoundRobin
        servers = [
            fixture_server("server1", rate=10, max_batch=16),
            fixture_server("server2", rate=10, max_batch=16),
            fixture_server("server3", cls="medium-priority", rate=10, max_batch=16),
        ]
        system, opt = greedy_system(
            servers=servers, policy="PriorityRoundRobin", delayed=True
        )
        solve_greedy(system, opt)
        assert len(allocated_servers(system)) >= 1

    def test_round_robin(self):
        # greedy_test.go:574 TestSolver_SolveGreedy_RoundRobin
        servers = [
            fixture_server("server1", rate=10, max_batch=16),
            fixture_server("server2", cls="medium-priority", rate=10, max_batch=16),
            fixture_server("server3", cls="low-priority", rate=10, max_batch=16),
        ]
        system, opt = greedy_system(servers=servers, policy="RoundRobin", delayed=True)
        solve_greedy(system, opt)
        assert len(allocated_servers(system)) >= 1

    def test_resource_exhaustion(self):
        # greedy_test.go:663 TestSolver_SolveGreedy_ResourceExhaustion:
        # 5 competing servers on a 1+1 pool — some must go unallocated,
        # at least one must win
        servers = [
            fixture_server(f"server{i}", rate=20, max_batch=16) for i in range(1, 6)
        ]
        system, opt = greedy_system(
            servers=servers,
            capacity=((MI300X_T, 1), (MI355X_T, 1)),
            policy="PriorityExhaustive",
            delayed=True,
        )
        solve_greedy(system, opt)
        n = len(allocated_servers(system))
        assert 1 <= n < 5
        # capacity respected in GPU units
        used = consumed_units(system)
        assert used.get(MI300X_T, 0) <= 1
        assert used.get(MI355X_T, 0) <= 1

    def test_high_load_scenario(self):
        # greedy_test.go:732 TestSolver_SolveGreedy_HighLoadScenario
        servers = [
            fixture_server("server1", rate=100, in_tok=200, out_tok=300,
                           min_replicas=2, max_batch=32),
            fixture_server("server2", cls="medium-priority", rate=80,
                           in_tok=150, out_tok=250, max_batch=16),
            fixture_server("server3", model="llama-70b", cls="low-priority",
                           rate=50, in_tok=200, out_tok=400, max_batch=8),
        ]
        system, opt = greedy_system(
            servers=servers, policy="PriorityExhaustive", delayed=True
        )
        solve_greedy(system, opt)
        assert len(allocated_servers(system)) >= 1

    def test_mixed_model_types(self):
        # greedy_test.go:828 TestSolver_SolveGreedy_MixedModelTypes
        servers = [
            fixture_server("llama8b-server", rate=40, max_batch=16),
            fixture_server("llama70b-server", model="llama-70b",
                           rate=30, in_tok=150, out_tok=300, max_batch=8),
        ]
        system, opt = greedy_system(servers=servers, policy="RoundRobin", delayed=True)
        solve_greedy(system, opt)
        assert len(allocated_servers(system)) >= 1

    def test_edge_cases_zero_and_extreme_load(self):
        # greedy_test.go:903 TestSolver_SolveGreedy_EdgeCases
        servers = [
            fixture_server("zero-load-server", rate=0, max_batch=16),
            fixture_server("high-load-server", cls="medium-priority", rate=1000,
                           in_tok=500, out_tok=1000, min_replicas=3, max_batch=64),
        ]
        system, opt = greedy_system(
            servers=servers, policy="PriorityRoundRobin", delayed=True
        )
        solve_greedy(system, opt)
        assert len(allocated_servers(system)) >= 1


class TestAllocateMaximallyEdgeCases:
    # greedy_test.go:979 TestAllocateMaximally_EdgeCases

    def test_empty_server_entries(self):
        system, _ = greedy_system()
        available = {MI300X_T: 4, MI355X_T: 2}
        _allocate_maximally(system, [], available)
        assert available == {MI300X_T: 4, MI355X_T: 2}

    def test_invalid_allocations(self):
        system, _ = greedy_system()
        available = {MI300X_T: 4, MI355X_T: 2}
        entries = [ServerEntry("nonexistent-server", 1, [])]
        _allocate_maximally(system, entries, available)
        assert available == {MI300X_T: 4, MI355X_T: 2}

    def test_no_available_resources(self):
        system, _ = greedy_system()
        available = {MI300X_T: 0, MI355X_T: 0}
        server = system.servers["server1"]
        original = server.allocation
        entry = first_entry(system, "server1")
        _allocate_maximally(system, [entry], available)
        assert server.allocation is original

    def test_maximal_allocation_consumes_resources(self):
        system, _ = greedy_system()
        available = {MI300X_T: 8, MI355X_T: 4}
        initial = dict(available)
        server = system.servers["server1"]
        server.remove_allocation()
        entry = first_entry(system, "server1", replicas=3)
        _allocate_maximally(system, [entry], available)
        if server.allocation is not None:
            assert any(available[t] < initial[t] for t in available)


class TestAllocateEquallyEdgeCases:
    # greedy_test.go:1115 TestAllocateEqually_EdgeCases

    def test_empty_server_entries(self):
        system, _ = greedy_system()
        available = {MI300X_T: 4, MI355X_T: 2}
        _allocate_equally(system, [], available)
        assert available == {MI300X_T: 4, MI355X_T: 2}

    def test_server_with_no_allocations(self):
        system, _ = greedy_system()
        available = {MI300X_T: 4, MI355X_T: 2}
        _allocate_equally(system, [ServerEntry("server1", 1, [])], available)
        assert available == {MI300X_T: 4, MI355X_T: 2}

    def test_round_robin_with_limited_resources(self):
        system, _ = greedy_system()
        available = {MI300X_T: 2, MI355X_T: 1}
        initial = dict(available)
        s1, s2 = system.servers["server1"], system.servers["server2"]
        s1.remove_allocation()
        s2.remove_allocation()
        entries = [
            first_entry(system, "server1", replicas=1),
            first_entry(system, "server2", replicas=1),
        ]
        _allocate_equally(system, entries, available)
        allocated = [s for s in (s1, s2) if s.allocation is not None]
        assert allocated, "at least one server allocated with resources available"
        for s in allocated:
            assert s.allocation.num_replicas > 0
        assert any(available[t] < initial[t] for t in available), (
            "resources consumed when allocations were made"
        )

    def test_multiple_round_robin_rounds(self):
        system, _ = greedy_system()
        available = {MI300X_T: 6, MI355X_T: 3}
        s1, s3 = system.servers["server1"], system.servers["server3"]
        s1.remove_allocation()
        s3.remove_allocation()
        entries = [
            first_entry(system, "server1", replicas=3),
            first_entry(system, "server3", replicas=3),
        ]
        _allocate_equally(system, entries, available)
        assert any(s.allocation is not None for s in (s1, s3))


class TestAllocateEquallyTicketManagement:
    # greedy_test.go:1309 TestAllocateEqually_TicketManagement

    def test_ticket_lifecycle(self):
        system, _ = greedy_system()
        available = {MI300X_T: 4, MI355X_T: 2}
        initial = dict(available)
        server = system.servers["server1"]
        server.remove_allocation()
        entry = first_entry(system, "server1", replicas=2)
        _allocate_equally(system, [entry], available)
        assert server.allocation is not None
        assert server.allocation.num_replicas > 0
        assert any(available[t] < initial[t] for t in available)

    def test_ticket_removal_on_resource_exhaustion(self):
        system, _ = greedy_system()
        available = {MI300X_T: 0, MI355X_T: 0}
        server = system.servers["server1"]
        server.remove_allocation()
        entry = first_entry(system, "server1", replicas=1)
        _allocate_equally(system, [entry], available)
        assert server.allocation is None


class TestBestEffortMatrix:
    # greedy_test.go:308 TestBestEffort_None + :1408 TestBestEffort

    def test_none_policy_leaves_available_untouched(self):
        system, _ = greedy_system()
        available = {MI300X_T: 4}
        _best_effort(system, [], available, "None")
        assert available == {MI300X_T: 4}

    def test_best_effort_with_multiple_entries(self):
        system, _ = greedy_system()
        available = {MI300X_T: 3, MI355X_T: 2}
        names = ["server1", "server2", "server3"]
        for n in names:
            system.servers[n].remove_allocation()
        entries = []
        for i, n in enumerate(names):
            e = first_entry(system, n, replicas=1)
            e.priority = i + 1
            entries.append(e)
        _best_effort(system, entries, available, "PriorityExhaustive")
        assert any(system.servers[n].allocation is not None for n in names)

    @pytest.mark.parametrize(
        "policy", ["PriorityRoundRobin", "RoundRobin", "None", "UnknownPolicy"]
    )
    def test_best_effort_with_different_policies(self, policy):
        system, _ = greedy_system()
        available = {MI300X_T: 2, MI355X_T: 1}
        server = system.servers["server1"]
        server.remove_allocation()
        entry = first_entry(system, "server1", replicas=1)
        _best_effort(system, [entry], available, policy)  # must not raise
        if policy == "None":
            assert server.allocation is None


class TestAllocateComprehensive:
    # greedy_test.go:1516 TestAllocate_ComprehensiveCoverage

    def test_empty_entries(self):
        system, _ = greedy_system()
        available = {MI300X_T: 4, MI355X_T: 2}
        assert _allocate(system, [], available) == []
        assert available == {MI300X_T: 4, MI355X_T: 2}

    def test_entries_with_no_allocations(self):
        system, _ = greedy_system()
        available = {MI300X_T: 4, MI355X_T: 2}
        entries = [ServerEntry("server1", 1, [])]
        assert _allocate(system, entries, available) == []

    def test_nonexistent_server_skipped(self):
        system, _ = greedy_system()
        available = {MI300X_T: 4, MI355X_T: 2}
        entries = [ServerEntry("nonexistent-server", 1, [])]
        assert _allocate(system, entries, available) == []
        assert available == {MI300X_T: 4, MI355X_T: 2}

    def test_resource_exhaustion_with_reordering(self):
        # the Go test's walk: miss -> advance curIndex + reinsert; after
        # the last candidate misses the entry lands in unallocated
        system, _ = greedy_system()
        available = {MI300X_T: 0, MI355X_T: 0}
        entry = first_entry(system, "server1", replicas=10, value=10.0, count=3)
        assert len(entry.allocations) >= 2
        unallocated = _allocate(system, [entry], available)
        assert len(unallocated) == 1
        assert unallocated[0].server_name == "server1"
        assert unallocated[0].cur_index == len(unallocated[0].allocations)


class TestMakePriorityGroupsMatrix:
    # greedy_test.go:331-409

    def test_empty_entries(self):
        assert make_priority_groups([]) == []

    def test_single_priority(self):
        entries = [ServerEntry(f"s{i}", 5, []) for i in range(3)]
        groups = make_priority_groups(entries)
        assert len(groups) == 1 and len(groups[0]) == 3

    def test_multiple_priorities(self):
        entries = (
            [ServerEntry("a", 1, [])]
            + [ServerEntry("b", 2, []), ServerEntry("c", 2, [])]
            + [ServerEntry("d", 7, [])]
        )
        groups = make_priority_groups(entries)
        assert [len(g) for g in groups] == [1, 2, 1]

    def test_order_preservation(self):
        entries = [
            ServerEntry("x", 1, []),
            ServerEntry("y", 1, []),
            ServerEntry("z", 1, []),
        ]
        groups = make_priority_groups(entries)
        assert [e.server_name for e in groups[0]] == ["x", "y", "z"]


class TestGreedyCapacityInvariants:
    """Cross-scenario invariant the Go suite checks piecemeal: whatever
    the policy, chosen allocations never exceed the capacity pool."""

    @pytest.mark.parametrize(
        "policy,delayed",
        [
            ("None", False),
            ("PriorityExhaustive", True),
            ("PriorityRoundRobin", True),
            ("RoundRobin", True),
            ("PriorityExhaustive", False),
            ("RoundRobin", False),
        ],
    )
    def test_capacity_never_exceeded(self, policy, delayed):
        servers = [
            fixture_server(f"s{i}", rate=25, max_batch=16,
                           cls=["high-priority", "medium-priority", "low-priority"][i % 3])
            for i in range(6)
        ]
        system, opt = greedy_system(
            servers=servers,
            capacity=((MI300X_T, 3), (MI355X_T, 2)),
            policy=policy,
            delayed=delayed,
        )
        solve_greedy(system, opt)
        used = consumed_units(system)
        assert used.get(MI300X_T, 0) <= 3, f"{policy}: {used}"
        assert used.get(MI355X_T, 0) <= 2, f"{policy}: {used}"

    def test_priority_strictness_under_scarcity(self):
        # the high-priority server must win the contested pool
        servers = [
            fixture_server("low", cls="low-priority", rate=25, max_batch=16),
            fixture_server("high", cls="high-priority", rate=25, max_batch=16),
        ]
        system, opt = greedy_system(
            servers=servers, capacity=((MI300X_T, 1), (MI355X_T, 0)), policy="None"
        )
        solve_greedy(system, opt)
        high, low = system.servers["high"], system.servers["low"]
        if high.allocation is None:
            # feasible-for-neither is the only excuse
            assert low.allocation is None


class TestRoundRobinEqualSplit:
    def test_two_equal_servers_split_the_pool(self):
        """Deterministic round-robin outcome: two identical servers each
        wanting 4 replicas on a 4-unit pool end with 2 each (ticket
        cycling grants one replica per turn)."""
        system, _ = greedy_system()
        available = {MI300X_T: 4, MI355X_T: 0}
        s1, s3 = system.servers["server1"], system.servers["server3"]
        s1.remove_allocation()
        s3.remove_allocation()
        # both llama-8b servers restricted to their MI300X candidate,
        # each wanting 4 replicas on a 4-unit pool
        e1 = ServerEntry("server1", 1, [
            a for n, a in s1.all_allocations.items() if n == "MI300X"
        ])
        e3 = ServerEntry("server3", 1, [
            a for n, a in s3.all_allocations.items() if n == "MI300X"
        ])
        assert e1.allocations and e3.allocations
        e1.allocations[0].num_replicas = 4
        e3.allocations[0].num_replicas = 4
        _allocate_equally(system, [e1, e3], available)
        a1, a3 = s1.allocation, s3.allocation
        assert a1 is not None and a3 is not None
        # identical unit weights: the round-robin tickets split evenly
        assert a1.num_replicas == 2 and a3.num_replicas == 2
        assert available[MI300X_T] == 0

    def test_single_server_gets_partial_grant(self):
        # wants 4 replicas, pool has 2 units -> best effort grants 2
        system, _ = greedy_system()
        available = {MI300X_T: 2, MI355X_T: 0}
        s1 = system.servers["server1"]
        s1.remove_allocation()
        e1 = ServerEntry("server1", 1, [
            a for n, a in s1.all_allocations.items() if n == "MI300X"
        ])
        e1.allocations[0].num_replicas = 4
        want_cost = e1.allocations[0].cost
        _allocate_equally(system, [e1], available)
        a1 = s1.allocation
        assert a1 is not None
        assert a1.num_replicas == 2
        assert available[MI300X_T] == 0
