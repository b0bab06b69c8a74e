"""Solver tests (mirrors pkg/solver/{solver,greedy,optimizer}_test.go:
unlimited argmin, greedy capacity constraints, priority ordering, regret
reordering on capacity miss, all four saturation policies, round-robin
ticket lifecycle, solve timing)."""

import pytest

from prometheus_client import CollectorRegistry

from wva_amd.config import OptimizerSpec
from wva_amd.controller import metrics as ctrl_metrics
from wva_amd.core import System
from wva_amd.solver import Manager, Optimizer, Solver
from wva_amd.solver.greedy import ServerEntry, make_priority_groups
from fixtures import make_system, server_spec


def solve(system, opt_spec):
    system.calculate()
    solver = Solver(opt_spec)
    solver.solve(system)
    return solver


class TestUnlimited:
    def test_argmin_value_per_server(self):
        system, opt = make_system(unlimited=True)
        solve(system, opt)
        server = system.server("s1:default")
        alloc = server.allocation
        assert alloc is not None
        best = min(server.all_allocations.values(), key=lambda a: a.value)
        assert alloc is best

    def test_prefers_cheaper_feasible_acc(self):
        # fresh server, no current allocation: value == cost; MI300X (65) is
        # cheaper than MI355X (85) at 1 replica each under light load
        system, opt = make_system(
            servers=[server_spec("s:ns", arrival_rate=30.0)], unlimited=True
        )
        solve(system, opt)
        alloc = system.server("s:ns").allocation
        assert alloc.accelerator == "MI300X"

    def test_keep_accelerator_sticks(self):
        system, opt = make_system(
            servers=[
                server_spec(
                    "s:ns",
                    keep_accelerator=True,
                    cur_accelerator="MI355X",
                    cur_replicas=1,
                    arrival_rate=30.0,
                )
            ],
            unlimited=True,
        )
        solve(system, opt)
        assert system.server("s:ns").allocation.accelerator == "MI355X"

    def test_diff_allocation(self):
        system, opt = make_system(
            servers=[server_spec("s:ns", cur_accelerator="MI355X", cur_replicas=5, arrival_rate=30.0)],
            unlimited=True,
        )
        solver = solve(system, opt)
        diff = solver.diff_allocation["s:ns"]
        assert diff.old_accelerator == "MI355X"
        assert diff.old_num_replicas == 5
        assert diff.new_num_replicas == system.server("s:ns").allocation.num_replicas


class TestGreedy:
    def test_respects_capacity(self):
        # one MI355X unit available; two Premium servers each needing >= 1
        system, opt = make_system(
            servers=[
                server_spec("a:ns", arrival_rate=60.0),
                server_spec("b:ns", arrival_rate=60.0),
            ],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 1), ("AMD-MI300X-192GB", 1)],
        )
        solve(system, opt)
        allocs = {n: s.allocation for n, s in system.servers.items()}
        accs = sorted(a.accelerator for a in allocs.values() if a is not None)
        # both got something, on distinct types
        assert accs == ["MI300X", "MI355X"]

    def test_priority_order_wins(self):
        # capacity for only one server; Premium (prio 1) must win over Freemium
        system, opt = make_system(
            servers=[
                server_spec("free:ns", class_name="Freemium", arrival_rate=60.0),
                server_spec("prem:ns", class_name="Premium", arrival_rate=60.0),
            ],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 1)],
        )
        # restrict to a single accelerator type pool
        system.remove_accelerator("MI300X")
        system.remove_accelerator("L40S")
        solve(system, opt)
        assert system.server("prem:ns").allocation is not None
        assert system.server("free:ns").allocation is None

    def test_exhausted_capacity_none_policy(self):
        system, opt = make_system(
            servers=[server_spec("a:ns", arrival_rate=60000.0)],  # needs many replicas
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 1), ("AMD-MI300X-192GB", 0)],
            saturation_policy="None",
        )
        solve(system, opt)
        assert system.server("a:ns").allocation is None

    def test_priority_exhaustive_gives_partial(self):
        system, opt = make_system(
            servers=[server_spec("a:ns", arrival_rate=60000.0)],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 2), ("AMD-MI300X-192GB", 0)],
            saturation_policy="PriorityExhaustive",
        )
        system.remove_accelerator("MI300X")
        system.remove_accelerator("L40S")
        solve(system, opt)
        alloc = system.server("a:ns").allocation
        assert alloc is not None
        assert alloc.num_replicas == 2  # all remaining capacity

    def test_round_robin_shares_capacity(self):
        system, opt = make_system(
            servers=[
                server_spec("a:ns", arrival_rate=60000.0),
                server_spec("b:ns", arrival_rate=60000.0),
            ],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 4)],
            saturation_policy="RoundRobin",
            delayed_best_effort=True,
        )
        system.remove_accelerator("MI300X")
        system.remove_accelerator("L40S")
        solve(system, opt)
        a = system.server("a:ns").allocation
        b = system.server("b:ns").allocation
        assert a is not None and b is not None
        assert a.num_replicas + b.num_replicas == 4
        assert abs(a.num_replicas - b.num_replicas) <= 1

    def test_best_effort_rescales_cost_and_value(self):
        system, opt = make_system(
            servers=[server_spec("a:ns", arrival_rate=60000.0)],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 2)],
            saturation_policy="PriorityExhaustive",
        )
        system.remove_accelerator("MI300X")
        system.remove_accelerator("L40S")
        system.calculate()
        desired = system.server("a:ns").all_allocations["MI355X"].num_replicas
        assert desired > 2
        solver = Solver(opt)
        solver.solve(system)
        alloc = system.server("a:ns").allocation
        assert alloc.cost == pytest.approx(85.0 * 2)

    def test_regret_reordering_on_capacity_miss(self):
        # two same-priority servers compete; one MI355X unit + one MI300X unit.
        # The one with larger regret (delta to next-best) must get MI355X first.
        system, opt = make_system(
            servers=[
                # current acc MI300X -> switching to MI355X is penalized, so
                # small delta between options
                server_spec("small-regret:ns", arrival_rate=60.0, cur_accelerator="MI300X", cur_replicas=1),
                # fresh server: value==cost, delta = 85-65 = 20
                server_spec("fresh:ns", arrival_rate=60.0),
            ],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 1), ("AMD-MI300X-192GB", 1)],
        )
        system.remove_accelerator("L40S")
        solve(system, opt)
        a = system.server("small-regret:ns").allocation
        b = system.server("fresh:ns").allocation
        assert a is not None and b is not None
        assert {a.accelerator, b.accelerator} == {"MI355X", "MI300X"}
        # small-regret keeps its current MI300X (cheapest by penalty), fresh
        # takes its own argmin (MI300X is cheaper, but only one unit) —
        # whoever sorts first gets its first choice; both end up allocated.


class TestPriorityGroups:
    def test_grouping(self):
        def e(p):
            entry = ServerEntry("s", p, [])
            return entry

        groups = make_priority_groups([e(1), e(1), e(5), e(10), e(10), e(10)])
        assert [len(g) for g in groups] == [2, 1, 3]
        assert [g[0].priority for g in groups] == [1, 5, 10]

    def test_empty(self):
        assert make_priority_groups([]) == []


class TestOptimizerAndManager:
    def test_solve_is_timed_and_observed(self):
        registry = CollectorRegistry()
        ctrl_metrics.init_metrics(registry)
        try:
            system, opt_spec = make_system()
            system.calculate()
            optimizer = Optimizer(opt_spec)
            manager = Manager(system, optimizer)
            manager.optimize()
            assert optimizer.solution_time_msec >= 0.0
            assert system.server("s1:default").allocation is not None
            # histogram observed one solve
            count = registry.get_sample_value("wva_solver_duration_seconds_count")
            assert count == 1.0
            # allocate_by_type ran
            assert system.allocation_by_type
        finally:
            ctrl_metrics.reset_metrics()

    def test_missing_spec_raises(self):
        optimizer = Optimizer(None)
        with pytest.raises(ValueError):
            optimizer.optimize(System())
