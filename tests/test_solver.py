"""Solver tests (mirrors pkg/solver/{solver,greedy,optimizer}_test.go:
unlimited argmin, greedy capacity constraints, priority ordering, regret
reordering on capacity miss, all four saturation policies, round-robin
ticket lifecycle, solve timing)."""

import pytest

from prometheus_client import CollectorRegistry

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


class TestGreedyDeep:
    """Deeper limited-mode coverage mirroring the reference's heavy greedy
    suite (pkg/solver/greedy_test.go, ~1.7k LoC)."""

    def _two_tier(self, cap_355, cap_300, policy="None", delayed=False, rates=(60.0, 60.0, 60.0)):
        system, opt = make_system(
            servers=[
                server_spec("p1:ns", class_name="Premium", arrival_rate=rates[0]),
                server_spec("p2:ns", class_name="Premium", arrival_rate=rates[1]),
                server_spec("f1:ns", class_name="Freemium", arrival_rate=rates[2]),
            ],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", cap_355), ("AMD-MI300X-192GB", cap_300)],
            saturation_policy=policy,
            delayed_best_effort=delayed,
        )
        system.remove_accelerator("L40S")
        return system, opt

    def test_priority_groups_allocated_in_order(self):
        # capacity for exactly two single-replica servers: both Premium win
        system, opt = self._two_tier(1, 1)
        solve(system, opt)
        assert system.server("p1:ns").allocation is not None
        assert system.server("p2:ns").allocation is not None
        assert system.server("f1:ns").allocation is None

    def test_lower_priority_gets_leftovers(self):
        system, opt = self._two_tier(2, 1)
        solve(system, opt)
        allocs = [system.server(n).allocation for n in ("p1:ns", "p2:ns", "f1:ns")]
        assert all(a is not None for a in allocs)

    def test_multiplicity_consumes_units(self):
        # multiplicity 2: each replica consumes 2 units of the type pool
        system, opt = make_system(
            servers=[server_spec("s:ns", arrival_rate=60.0)],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 2)],
        )
        system.remove_accelerator("MI300X")
        system.remove_accelerator("L40S")
        system.accelerator("MI355X").spec.multiplicity = 2
        solve(system, opt)
        alloc = system.server("s:ns").allocation
        assert alloc is not None and alloc.num_replicas == 1
        system.allocate_by_type()
        assert system.allocation_by_type["AMD-MI355X-288GB"].count == 2

    def test_acc_count_consumes_units(self):
        # llama-70b uses 4 MI355X instances per replica: capacity 4 fits 1
        system, opt = make_system(
            servers=[server_spec("s:ns", model="llama-70b", arrival_rate=60.0)],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 4)],
        )
        system.remove_accelerator("MI300X")
        system.remove_accelerator("L40S")
        solve(system, opt)
        alloc = system.server("s:ns").allocation
        assert alloc is not None and alloc.accelerator == "MI355X"

    def test_round_robin_pours_out_all_capacity(self):
        # saturation round-robin keeps granting replicas while capacity
        # remains (greedy.go:293-298 quirk: no cap at the desired count)
        system, opt = make_system(
            servers=[
                server_spec("a:ns", arrival_rate=60000.0),
                server_spec("b:ns", arrival_rate=60000.0),
            ],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 10)],
            saturation_policy="RoundRobin",
            delayed_best_effort=True,
        )
        system.remove_accelerator("MI300X")
        system.remove_accelerator("L40S")
        solve(system, opt)
        a = system.server("a:ns").allocation
        b = system.server("b:ns").allocation
        assert a.num_replicas + b.num_replicas == 10
        assert abs(a.num_replicas - b.num_replicas) <= 1

    def test_priority_round_robin_groups(self):
        system, opt = self._two_tier(
            0, 0, policy="PriorityRoundRobin", rates=(60000.0, 60000.0, 60000.0)
        )
        system.capacity["AMD-MI355X-288GB"] = 4
        solve(system, opt)
        # Premium group shares the pool round-robin before Freemium sees it
        p1 = system.server("p1:ns").allocation
        p2 = system.server("p2:ns").allocation
        assert p1 is not None and p2 is not None
        assert p1.num_replicas + p2.num_replicas == 4
        assert system.server("f1:ns").allocation is None

    def test_delayed_best_effort_spans_priorities(self):
        # delayed mode: best effort runs once at the end over ALL leftover
        # servers, so a Freemium server can still receive capacity that
        # per-group mode would have burned inside the Premium group
        system, opt = self._two_tier(
            0, 0, policy="RoundRobin", delayed=True, rates=(60000.0, 60000.0, 60.0)
        )
        system.capacity["AMD-MI355X-288GB"] = 5
        solve(system, opt)
        got = [
            system.server(n).allocation is not None for n in ("p1:ns", "p2:ns", "f1:ns")
        ]
        assert got.count(True) >= 2  # capacity reached beyond one group

    def test_regret_delta_updates_on_miss(self):
        from wva_amd.solver.greedy import ServerEntry, _allocate

        system, opt = make_system(
            servers=[server_spec("s:ns", arrival_rate=60.0)],
            unlimited=False,
            capacity=[("AMD-MI355X-288GB", 0), ("AMD-MI300X-192GB", 1)],
        )
        system.remove_accelerator("L40S")
        system.calculate()
        server = system.server("s:ns")
        allocs = sorted(server.all_allocations.values(), key=lambda a: a.value)
        entry = ServerEntry("s:ns", 1, allocs)
        entry.delta = allocs[1].value - allocs[0].value
        unallocated = _allocate(system, [entry], dict(system.capacity))
        # first choice (cheaper MI300X? capacity says MI355X empty) —
        # whichever missed, the entry advanced and was retried, ending
        # allocated on the type with capacity
        assert unallocated == []
        assert server.allocation is not None
        assert server.allocation.accelerator == "MI300X"


class TestEnergyObjective:
    """cost+energy objective: the MI355X power curve enters the value
    (extension — the reference computes Power() but never uses it)."""

    def _system(self, price, arrival=30.0):
        system, opt = make_system(
            servers=[server_spec("s:ns", arrival_rate=arrival)], unlimited=True
        )
        opt.objective = "cost+energy"
        opt.energy_cost_per_kwh = price
        system.optimizer_spec = opt
        return system, opt

    def test_zero_price_is_noop(self):
        base_sys, base_opt = make_system(
            servers=[server_spec("s:ns", arrival_rate=30.0)], unlimited=True
        )
        solve(base_sys, base_opt)
        e_sys, e_opt = self._system(price=0.0)
        solve(e_sys, e_opt)
        for acc in base_sys.server("s:ns").all_allocations:
            assert base_sys.server("s:ns").all_allocations[acc].value == pytest.approx(
                e_sys.server("s:ns").all_allocations[acc].value
            )

    def test_energy_term_added_to_value(self):
        system, opt = self._system(price=10.0)  # 10 cents/kWh
        solve(system, opt)
        server = system.server("s:ns")
        alloc = server.all_allocations["MI355X"]
        acc = system.accelerator("MI355X")
        base = server.cur_allocation.transition_penalty(alloc)
        expected = base + acc.power(alloc.rho) * alloc.num_replicas / 1000.0 * 10.0
        assert alloc.value == pytest.approx(expected)

    def test_high_energy_price_flips_choice(self):
        # MI300X is cheaper in unit cost AND power; under pure cost the
        # light-load argmin already picks MI300X — so craft the flip the
        # other way: make MI355X cheaper in cost but thirstier, and verify
        # a high energy price moves the argmin to the frugal card
        system, opt = self._system(price=0.0, arrival=30.0)
        system.accelerator("MI355X").spec.cost = 60.0  # cheaper than MI300X@65
        for srv in system.servers.values():
            srv.max_batch_size = 8
        solve(system, opt)
        assert system.server("s:ns").allocation.accelerator == "MI355X"

        # near-idle both cards draw ~idle power (MI355X ~200 W vs MI300X
        # ~160 W at this load), so a large price difference is needed to
        # overcome the 5.5-cent value gap
        system2, opt2 = self._system(price=1000.0, arrival=30.0)
        system2.accelerator("MI355X").spec.cost = 60.0
        solve(system2, opt2)
        assert system2.server("s:ns").allocation.accelerator == "MI300X"

    def test_batched_path_matches_scalar(self):
        from wva_amd.ops import BatchedAllocationSolver

        scalar_sys, opt = self._system(price=25.0)
        solve(scalar_sys, opt)
        batch_sys, opt2 = self._system(price=25.0)
        for g in batch_sys.accelerators.values():
            g.calculate()
        BatchedAllocationSolver().calculate(batch_sys)
        for acc, a in scalar_sys.server("s:ns").all_allocations.items():
            b = batch_sys.server("s:ns").all_allocations[acc]
            assert b.value == pytest.approx(a.value, rel=1e-6)


class TestSolveCLI:
    def test_solve_spec_roundtrip(self, tmp_path):
        import json
        import subprocess
        import sys as _sys

        from fixtures import make_spec

        spec = make_spec(
            servers=[server_spec("a:ns", arrival_rate=600.0), server_spec("b:ns", arrival_rate=60.0)]
        )
        path = tmp_path / "system.json"
        path.write_text(json.dumps({"system": spec.to_dict()}))
        proc = subprocess.run(
            [_sys.executable, "-m", "wva_amd.solve", str(path)],
            capture_output=True,
            text=True,
            timeout=120,
            cwd=str(__import__("pathlib").Path(__file__).resolve().parent.parent),
        )
        assert proc.returncode == 0, proc.stderr
        out = json.loads(proc.stdout)
        assert set(out["allocations"]) == {"a:ns", "b:ns"}
        assert out["solutionTimeMsec"] >= 0
        assert out["allocations"]["a:ns"]["numReplicas"] >= 1
        assert out["unallocated"] == []

    def test_bare_spec_accepted(self):
        from wva_amd.solve import load_spec, solve_spec
        from fixtures import make_spec

        spec = make_spec()
        result = solve_spec(load_spec(spec.to_json()))
        assert "s1:default" in result["allocations"]


class TestManagerEdgeCases:
    """manager_test.go:334 TestManager_EdgeCases — invalid wiring must
    surface as an error, never silently no-op."""

    def test_nil_system_errors(self):
        from wva_amd.config import OptimizerSpec

        manager = Manager(None, Optimizer(OptimizerSpec(unlimited=True)))
        with pytest.raises(Exception):
            manager.optimize()

    def test_nil_optimizer_errors(self):
        system, _ = make_system()
        manager = Manager(system, None)
        with pytest.raises(Exception):
            manager.optimize()

    def test_optimize_produces_solution_and_type_aggregation(self):
        # manager_test.go:61 TestManager_Optimize happy path: optimize,
        # then the per-type aggregation and solution export line up
        system, opt = make_system(
            servers=[server_spec("m1:ns", arrival_rate=120.0)]
        )
        system.calculate()
        Manager(system, Optimizer(opt)).optimize()
        solution = system.generate_solution()
        assert "m1:ns" in solution.spec
        entry = solution.spec["m1:ns"]
        assert entry.num_replicas >= 1
        by_type = system.allocation_by_type
        total_units = sum(a.count for a in by_type.values())
        assert total_units >= entry.num_replicas
