"""Core domain tests (mirrors pkg/core/*_test.go: spec round-trips,
allocation sizing, transition penalties, saturation, power curve)."""

import math

import pytest

from wva_amd.core import Accelerator, Allocation, System, create_allocation
from fixtures import MI355X, make_system, server_spec


class TestAccelerator:
    def test_power_curve(self):
        acc = Accelerator(MI355X)
        acc.calculate()
        assert acc.power(0.0) == pytest.approx(140.0)
        assert acc.power(0.6) == pytest.approx(900.0)
        assert acc.power(1.0) == pytest.approx(1400.0)
        # piecewise linear midpoints
        assert acc.power(0.3) == pytest.approx(140 + (900 - 140) / 0.6 * 0.3)
        assert acc.power(0.8) == pytest.approx(900 + (1400 - 900) / 0.4 * 0.2)


class TestServiceClass:
    def test_priority_clamping(self):
        system, _ = make_system()
        system.add_service_class("weird", 1000)
        assert system.service_class("weird").priority == 100
        system.add_service_class("neg", 0)
        assert system.service_class("neg").priority == 100


class TestCreateAllocation:
    def test_feasible_allocation(self):
        system, _ = make_system()
        system.calculate()
        alloc = create_allocation(system, "s1:default", "MI355X")
        assert alloc is not None
        assert alloc.accelerator == "MI355X"
        assert alloc.num_replicas >= 1
        assert alloc.cost == pytest.approx(85.0 * alloc.num_replicas)
        assert alloc.itl <= 20.0 * 1.001  # meets Premium SLO
        assert alloc.max_arrv_rate_per_replica > 0

    def test_unknown_names_return_none(self):
        system, _ = make_system()
        assert create_allocation(system, "nope", "MI355X") is None
        assert create_allocation(system, "s1:default", "nope") is None

    def test_infeasible_slo_returns_none(self):
        # L40S alpha=20 > Premium ITL target of 20*... below alpha+beta
        system, _ = make_system()
        alloc = create_allocation(system, "s1:default", "L40S")
        assert alloc is None

    def test_replica_scaling_with_load(self):
        sys_lo, _ = make_system(servers=[server_spec("s:ns", arrival_rate=60.0)])
        sys_hi, _ = make_system(servers=[server_spec("s:ns", arrival_rate=6000.0)])
        lo = create_allocation(sys_lo, "s:ns", "MI355X")
        hi = create_allocation(sys_hi, "s:ns", "MI355X")
        assert hi.num_replicas > lo.num_replicas
        # replica count = ceil(totalRate / rate*)
        total_rate = 6000.0 / 60.0
        expected = max(
            math.ceil(total_rate / (hi.max_arrv_rate_per_replica * 1000.0)), 1
        )
        assert hi.num_replicas == expected

    def test_zero_load_min_replicas(self):
        system, _ = make_system(servers=[server_spec("s:ns", arrival_rate=0.0, min_replicas=1)])
        alloc = create_allocation(system, "s:ns", "MI355X")
        assert alloc.num_replicas == 1
        assert alloc.rho == 0.0
        assert alloc.cost == pytest.approx(85.0)

    def test_zero_load_scale_to_zero(self):
        system, _ = make_system(servers=[server_spec("s:ns", arrival_rate=0.0, min_replicas=0)])
        alloc = create_allocation(system, "s:ns", "MI355X")
        assert alloc.num_replicas == 0
        assert alloc.accelerator == ""
        assert alloc.cost == 0.0

    def test_acc_count_multiplies_cost(self):
        system, _ = make_system(
            servers=[server_spec("s:ns", model="llama-70b", arrival_rate=600.0, max_batch=8)]
        )
        alloc = create_allocation(system, "s:ns", "MI355X")
        assert alloc is not None
        # llama-70b on MI355X uses acc_count=4 instances per replica
        assert alloc.cost == pytest.approx(85.0 * 4 * alloc.num_replicas)


class TestTransitionPenalty:
    def test_same_acc_same_replicas(self):
        a = Allocation(accelerator="MI355X", num_replicas=2, cost=170.0)
        b = Allocation(accelerator="MI355X", num_replicas=2, cost=170.0)
        assert a.transition_penalty(b) == 0.0

    def test_same_acc_scaling(self):
        a = Allocation(accelerator="MI355X", num_replicas=2, cost=170.0)
        b = Allocation(accelerator="MI355X", num_replicas=3, cost=255.0)
        assert a.transition_penalty(b) == pytest.approx(85.0)

    def test_acc_change_penalized(self):
        a = Allocation(accelerator="MI355X", num_replicas=2, cost=170.0)
        b = Allocation(accelerator="MI300X", num_replicas=2, cost=130.0)
        assert a.transition_penalty(b) == pytest.approx(0.1 * (170 + 130) + (130 - 170))


class TestSaturation:
    def test_saturated(self):
        a = Allocation(accelerator="MI355X", num_replicas=1, max_arrv_rate_per_replica=0.001)
        # maxRPM = 0.001 * 1000 * 60 = 60 req/min
        assert not a.saturated(59.0)
        assert a.saturated(61.0)


class TestServer:
    def test_keep_accelerator_pins_candidates(self):
        system, _ = make_system(
            servers=[server_spec("s:ns", keep_accelerator=True, cur_accelerator="MI300X", cur_replicas=1)]
        )
        system.calculate()
        server = system.server("s:ns")
        assert set(server.all_allocations) == {"MI300X"}

    def test_no_pin_enumerates_all_feasible(self):
        system, _ = make_system()
        system.calculate()
        server = system.server("s1:default")
        # L40S infeasible under Premium SLO; MI355X and MI300X feasible
        assert set(server.all_allocations) == {"MI355X", "MI300X"}

    def test_value_is_transition_penalty(self):
        system, _ = make_system(
            servers=[server_spec("s:ns", cur_accelerator="MI355X", cur_replicas=1)]
        )
        system.calculate()
        server = system.server("s:ns")
        alloc = server.all_allocations["MI355X"]
        cur = server.cur_allocation
        assert alloc.value == pytest.approx(cur.transition_penalty(alloc))


class TestSystem:
    def test_allocate_by_type_and_solution(self):
        system, _ = make_system(
            servers=[
                server_spec("a:ns", arrival_rate=600.0),
                server_spec("b:ns", model="llama-70b", arrival_rate=600.0),
            ]
        )
        system.calculate()
        for server in system.servers.values():
            server.set_allocation(server.all_allocations["MI355X"])
        system.allocate_by_type()
        t = system.allocation_by_type["AMD-MI355X-288GB"]
        a = system.server("a:ns").allocation
        b = system.server("b:ns").allocation
        assert t.count == a.num_replicas * 1 + b.num_replicas * 4
        assert t.cost == pytest.approx(a.cost + b.cost)

        sol = system.generate_solution()
        assert set(sol.spec) == {"a:ns", "b:ns"}
        assert sol.spec["a:ns"].accelerator == "MI355X"
        assert sol.spec["a:ns"].load.arrival_rate == 600.0

    def test_registry_removal(self):
        system, _ = make_system()
        system.remove_accelerator("L40S")
        assert system.accelerator("L40S") is None
        with pytest.raises(KeyError):
            system.remove_accelerator("L40S")

    def test_registry_removal_full_surface(self):
        # system_test.go:570/923/1179 — model / service-class / capacity
        # removal; missing model and class raise, missing capacity no-ops
        system, _ = make_system()
        model_name = next(iter(system.models))
        system.remove_model(model_name)
        assert system.model(model_name) is None
        with pytest.raises(KeyError):
            system.remove_model(model_name)

        cls_name = next(iter(system.service_classes))
        system.remove_service_class(cls_name)
        assert system.service_class(cls_name) is None
        with pytest.raises(KeyError):
            system.remove_service_class(cls_name)

        if system.capacity:
            t = next(iter(system.capacity))
            system.remove_capacity(t)
            assert t not in system.capacity
        system.remove_capacity("ghost-type")  # no-op


class TestScaleAndReallocate:
    def test_scale_tracks_load_change(self):
        from wva_amd.core.allocation import scale_allocation

        system, _ = make_system(servers=[server_spec("s:ns", arrival_rate=600.0)])
        alloc = create_allocation(system, "s:ns", "MI355X")
        system.server("s:ns").load.arrival_rate = 6000.0
        new_alloc, inc = scale_allocation(system, alloc, "s:ns")
        assert new_alloc is not None
        assert inc == new_alloc.num_replicas - alloc.num_replicas > 0

    def test_reallocate_picks_min_value(self):
        from wva_amd.core.allocation import reallocate

        system, _ = make_system(servers=[server_spec("s:ns", arrival_rate=30.0)])
        alloc, acc = reallocate(system, "s:ns")
        assert alloc is not None
        # cheapest feasible accelerator wins (fresh server: value == cost)
        assert acc == "MI300X"

    def test_reallocate_infeasible(self):
        from wva_amd.core.allocation import reallocate

        system, _ = make_system()
        alloc, acc = reallocate(system, "missing")
        assert alloc is None and acc == ""


class TestDefaultServiceClass:
    def test_empty_class_name_defaults_to_free(self):
        from wva_amd.config import DEFAULT_SERVICE_CLASS_NAME

        system, _ = make_system(servers=[server_spec("s:ns", class_name="")])
        server = system.server("s:ns")
        assert server.service_class_name == DEFAULT_SERVICE_CLASS_NAME == "Free"
        # no "Free" class registered -> no feasible allocation (parity with
        # GetServiceClass nil -> CreateAllocation nil)
        assert create_allocation(system, "s:ns", "MI355X") is None
        # priority falls back to the default lowest
        assert server.priority(system) == 100


class TestKVCacheCapacity:
    """MI355X HBM capacity model (wva_amd/core/kvcache.py)."""

    def _llama8b(self):
        from wva_amd.core.kvcache import ModelMemoryProfile

        return ModelMemoryProfile.from_architecture(
            params_billions=8, layers=32, kv_heads=8, head_dim=128
        )

    def test_kv_bytes_per_token_gqa(self):
        from wva_amd.core.kvcache import kv_bytes_per_token

        # Llama-3.1-8B GQA: 2 * 32 * 8 * 128 * 2 = 128 KiB
        assert kv_bytes_per_token(32, 8, 128, 2) == 128 * 1024
        with pytest.raises(ValueError):
            kv_bytes_per_token(0, 8, 128)

    def test_mi355x_8b_capacity(self):
        from wva_amd.core.kvcache import max_batch_for_context, max_concurrent_tokens

        profile = self._llama8b()
        tokens = max_concurrent_tokens(288, profile)
        assert 1_900_000 < tokens < 2_100_000  # ~2M tokens beside 16 GB weights
        assert 450 <= max_batch_for_context(288, profile, 4096) <= 520

    def test_weights_do_not_fit(self):
        from wva_amd.core.kvcache import ModelMemoryProfile, max_concurrent_tokens

        huge = ModelMemoryProfile.from_architecture(500, 120, 8, 128)  # 1 TB bf16
        assert max_concurrent_tokens(288, huge) == 0

    def test_fp8_doubles_capacity(self):
        from wva_amd.core.kvcache import ModelMemoryProfile, max_concurrent_tokens

        bf16 = self._llama8b()
        fp8 = ModelMemoryProfile.from_architecture(8, 32, 8, 128, dtype_bytes=1)
        # both weights and KV halve -> more than 2x the token budget
        assert max_concurrent_tokens(288, fp8) > 2 * max_concurrent_tokens(288, bf16)

    def test_validate_max_batch(self):
        from wva_amd.core.kvcache import validate_max_batch

        profile = self._llama8b()
        assert validate_max_batch(256, 288, profile, 4096) == ""
        warn = validate_max_batch(4096, 288, profile, 4096)
        assert "exceeds KV capacity" in warn and "4096" in warn

    def test_input_validation(self):
        from wva_amd.core.kvcache import max_batch_for_context, max_concurrent_tokens

        profile = self._llama8b()
        with pytest.raises(ValueError):
            max_concurrent_tokens(0, profile)
        with pytest.raises(ValueError):
            max_concurrent_tokens(288, profile, overhead_fraction=1.0)
        with pytest.raises(ValueError):
            max_batch_for_context(288, profile, 0)


class TestAtTokensFallbackQuirk:
    """De-facto contract from the reference (allocation.go:77-87 vs
    internal/utils/utils.go:218-232): when the server carries no
    maxBatchSize override, N falls back to max(perf.maxBatchSize *
    atTokens / K, 1) — and the controller path never sets atTokens, so
    the fallback degenerates to N=1, making the VA maxBatchSize
    effectively mandatory."""

    def _system(self, at_tokens):
        from tests.fixtures import make_system, server_spec

        system, _ = make_system(servers=[server_spec("s:ns", max_batch=0, out_tokens=128)])
        perf = system.model("llama-8b").get_perf_data("MI355X")
        perf.at_tokens = at_tokens
        return system

    def _explicit(self, max_batch):
        from tests.fixtures import make_system, server_spec

        system, _ = make_system(servers=[server_spec("s:ns", max_batch=max_batch, out_tokens=128)])
        return system

    def test_controller_path_degenerates_to_batch_1(self):
        got = create_allocation(self._system(0), "s:ns", "MI355X")
        want = create_allocation(self._explicit(1), "s:ns", "MI355X")
        assert got is not None
        assert got.num_replicas == want.num_replicas
        assert got.itl == want.itl
        assert got.ttft == want.ttft

    def test_at_tokens_scales_derived_batch(self):
        # N = perf.maxBatchSize * atTokens // K with atTokens=4*K (K=128)
        system = self._system(4 * 128)
        expect_n = system.model("llama-8b").get_perf_data("MI355X").max_batch_size * 4
        got = create_allocation(system, "s:ns", "MI355X")
        assert got.batch_size == expect_n
        want = create_allocation(self._explicit(expect_n), "s:ns", "MI355X")
        assert got is not None
        assert got.num_replicas == want.num_replicas
        assert got.itl == want.itl
        assert got.max_arrv_rate_per_replica == want.max_arrv_rate_per_replica
