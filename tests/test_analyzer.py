"""Unit tests for wva_amd.analyzer.

Mirrors the reference test strategy:
/root/reference/pkg/analyzer/{queueanalyzer,queuemodel,utils}_test.go —
builder validation, prefill/decode formulas, Analyze, Size, effective
concurrency, M/M/1/K probability sums, a Little's-law property check, and
binary-search boundary classification/precision.
"""

import numpy as np
import pytest

from wva_amd.analyzer import (
    AboveRegion,
    AnalyzerError,
    BelowRegion,
    Configuration,
    DecodeParms,
    InRegion,
    MM1KModel,
    MM1ModelStateDependent,
    PrefillParms,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    TargetPerf,
    binary_search,
    effective_concurrency,
    within_tolerance,
)
from wva_amd.analyzer.queueanalyzer import build_service_rates


def make_analyzer(
    alpha=6.958,
    beta=0.042,
    gamma=20.0,
    delta=0.1,
    max_batch=8,
    max_queue=80,
    in_tokens=128,
    out_tokens=128,
):
    cfg = Configuration(
        max_batch_size=max_batch,
        max_queue_size=max_queue,
        service_parms=ServiceParms(
            prefill=PrefillParms(gamma=gamma, delta=delta),
            decode=DecodeParms(alpha=alpha, beta=beta),
        ),
    )
    return QueueAnalyzer(cfg, RequestSize(avg_input_tokens=in_tokens, avg_output_tokens=out_tokens))


# ---------------------------------------------------------------- queue models
class TestMM1K:
    def test_probability_sum_is_one(self):
        m = MM1KModel(K=10)
        m.solve(0.5, 1.0)
        assert m.is_valid
        assert m.p.sum() == pytest.approx(1.0, rel=1e-12)

    def test_rho_one_uniform(self):
        m = MM1KModel(K=4)
        m.solve(1.0, 1.0)
        assert m.is_valid
        assert np.allclose(m.p, 1.0 / 5)

    def test_invalid_inputs(self):
        m = MM1KModel(K=4)
        m.solve(-1.0, 1.0)
        assert not m.is_valid
        m.solve(1.0, 0.0)
        assert not m.is_valid

    def test_known_mm1k_stats(self):
        # M/M/1/2 with rho=0.5: p = [4/7, 2/7, 1/7]
        m = MM1KModel(K=2)
        m.solve(0.5, 1.0)
        assert np.allclose(m.p, [4 / 7, 2 / 7, 1 / 7])
        assert m.throughput == pytest.approx(0.5 * (1 - 1 / 7))
        n = 0 * 4 / 7 + 1 * 2 / 7 + 2 * 1 / 7
        assert m.avg_num_in_system == pytest.approx(n)
        assert m.avg_resp_time == pytest.approx(n / m.throughput)


class TestStateDependent:
    def test_probability_sum_and_positivity(self):
        sr = [0.1, 0.15, 0.18, 0.2]
        m = MM1ModelStateDependent(K=40, serv_rate=sr)
        m.solve(0.12, 1.0)
        assert m.is_valid
        assert m.p.sum() == pytest.approx(1.0, rel=1e-12)
        assert (m.p >= 0).all()

    def test_matches_direct_recursion(self):
        # compare log-space softmax against the naive product-form recursion
        sr = [0.08, 0.12, 0.15]
        K = 12
        lam = 0.1
        m = MM1ModelStateDependent(K=K, serv_rate=sr)
        m.solve(lam, 1.0)
        p = np.ones(K + 1)
        for n in range(K):
            mu = sr[min(n, len(sr) - 1)]
            p[n + 1] = p[n] * lam / mu
        p /= p.sum()
        assert np.allclose(m.p, p, rtol=1e-10)

    def test_no_overflow_for_large_k_high_load(self):
        # the reference needs MaxFloat rescale loops here; log space just works
        sr = [1e-4] * 512
        m = MM1ModelStateDependent(K=512 * 11, serv_rate=sr)
        m.solve(0.5, 1.0)  # lambda/mu = 5000 per state
        assert m.is_valid
        assert m.p.sum() == pytest.approx(1.0, rel=1e-9)
        assert np.isfinite(m.p).all()
        # almost all mass at the boundary state
        assert m.p[-1] > 0.99

    def test_littles_law(self):
        # property check mirroring queuemodel_test.go:498: N = X * T
        sr = list(np.linspace(0.05, 0.2, 16))
        m = MM1ModelStateDependent(K=160, serv_rate=sr)
        for lam in (0.01, 0.05, 0.1, 0.19):
            m.solve(lam, 1.0)
            assert m.is_valid
            assert m.avg_num_in_system == pytest.approx(
                m.throughput * m.avg_resp_time, rel=1e-9
            )
            # in-service version of Little's law
            assert m.avg_num_in_servers == pytest.approx(
                m.throughput * m.avg_serv_time, rel=1e-9
            )

    def test_zero_lambda(self):
        m = MM1ModelStateDependent(K=10, serv_rate=[0.1])
        m.solve(0.0, 1.0)
        assert m.is_valid
        assert m.p[0] == pytest.approx(1.0)
        assert m.throughput == pytest.approx(0.0)


# ---------------------------------------------------------------- binary search
class TestBinarySearch:
    def test_within_tolerance(self):
        assert within_tolerance(1.0, 1.0, 0.0)
        assert within_tolerance(1.0000005, 1.0, 1e-6)
        assert not within_tolerance(1.1, 1.0, 1e-6)
        assert not within_tolerance(0.1, 0.0, 1e-6)
        assert not within_tolerance(1.0, 2.0, -1.0)

    def test_increasing(self):
        x, ind = binary_search(0.0, 10.0, 25.0, lambda x: x * x)
        assert ind == InRegion
        assert x == pytest.approx(5.0, rel=1e-5)

    def test_decreasing(self):
        x, ind = binary_search(1.0, 10.0, 0.5, lambda x: 1.0 / x)
        assert ind == InRegion
        assert x == pytest.approx(2.0, rel=1e-5)

    def test_below_region(self):
        x, ind = binary_search(1.0, 10.0, 0.5, lambda x: x)
        assert ind == BelowRegion
        assert x == 1.0

    def test_above_region(self):
        x, ind = binary_search(1.0, 10.0, 50.0, lambda x: x)
        assert ind == AboveRegion
        assert x == 10.0

    def test_boundary_hit(self):
        x, ind = binary_search(1.0, 10.0, 1.0, lambda x: x)
        assert ind == InRegion
        assert x == 1.0

    def test_invalid_range(self):
        with pytest.raises(ValueError):
            binary_search(2.0, 1.0, 0.0, lambda x: x)


# ---------------------------------------------------------------- analyzer
class TestBuildModel:
    def test_service_rates_shape_and_values(self):
        qa = make_analyzer(alpha=10.0, beta=1.0, gamma=5.0, delta=0.01, max_batch=4, in_tokens=100, out_tokens=10)
        # mu(n) = n / (prefill(n) + 9*decode(n))
        for i, n in enumerate(range(1, 5)):
            prefill = 5.0 + 0.01 * 100 * n
            decode = 9 * (10.0 + 1.0 * n)
            assert qa.serv_rate[i] == pytest.approx(n / (prefill + decode))

    def test_decode_only_single_token(self):
        cfg = Configuration(
            max_batch_size=2,
            max_queue_size=20,
            service_parms=ServiceParms(
                prefill=PrefillParms(gamma=5.0, delta=0.1),
                decode=DecodeParms(alpha=10.0, beta=1.0),
            ),
        )
        rs = RequestSize(avg_input_tokens=0, avg_output_tokens=1)
        sr = build_service_rates(cfg, rs)
        # numDecode forced to 1, prefill contributes 0
        assert sr[0] == pytest.approx(1.0 / 11.0)
        assert sr[1] == pytest.approx(2.0 / 12.0)

    def test_rate_range(self):
        qa = make_analyzer()
        assert qa.rate_range.min == pytest.approx(float(qa.serv_rate[0]) * 0.001 * 1000)
        assert qa.rate_range.max == pytest.approx(float(qa.serv_rate[-1]) * 0.999 * 1000)

    def test_invalid_config(self):
        with pytest.raises(AnalyzerError):
            make_analyzer(max_batch=0)
        with pytest.raises(AnalyzerError):
            make_analyzer(max_queue=-1)
        with pytest.raises(AnalyzerError):
            make_analyzer(out_tokens=0)


class TestAnalyze:
    def test_basic_metrics(self):
        qa = make_analyzer()
        rate = qa.rate_range.max * 0.5
        m = qa.analyze(rate)
        assert 0 < m.throughput <= rate + 1e-9
        assert m.avg_resp_time > 0
        assert m.avg_wait_time >= 0
        assert 0 <= m.rho <= 1
        assert m.max_rate == pytest.approx(qa.rate_range.max)

    def test_rejects_bad_rates(self):
        qa = make_analyzer()
        with pytest.raises(AnalyzerError):
            qa.analyze(0.0)
        with pytest.raises(AnalyzerError):
            qa.analyze(qa.rate_range.max * 1.5)

    def test_monotone_wait_time(self):
        qa = make_analyzer()
        rates = np.linspace(qa.rate_range.min, qa.rate_range.max, 10)
        waits = [qa.analyze(float(r)).avg_wait_time for r in rates]
        assert all(b >= a - 1e-9 for a, b in zip(waits, waits[1:]))


class TestSize:
    def test_itl_target_inversion(self):
        qa = make_analyzer(alpha=6.958, beta=0.042, max_batch=64, max_queue=640)
        target = TargetPerf(target_itl=9.0)
        target_rate, metrics, achieved = qa.size(target)
        # achieved ITL must be at (or tolerably near) the target
        assert achieved.target_itl <= 9.0 * (1 + 1e-3)
        assert target_rate.rate_target_itl <= qa.rate_range.max

    def test_ttft_target_inversion(self):
        qa = make_analyzer(gamma=5.2, delta=0.1, in_tokens=64, max_batch=16, max_queue=160)
        target = TargetPerf(target_ttft=1000.0)
        target_rate, metrics, achieved = qa.size(target)
        assert achieved.target_ttft <= 1000.0 * (1 + 1e-3)

    def test_tps_bypasses_search(self):
        qa = make_analyzer()
        target_rate, _, _ = qa.size(TargetPerf(target_tps=100.0))
        assert target_rate.rate_target_tps == pytest.approx(qa.rate_range.max * 0.9)

    def test_infeasible_target_raises(self):
        # ITL target below alpha can never be met
        qa = make_analyzer(alpha=10.0, beta=0.5)
        with pytest.raises(AnalyzerError):
            qa.size(TargetPerf(target_itl=1.0))

    def test_loose_targets_give_lambda_max(self):
        qa = make_analyzer(max_batch=4, max_queue=40)
        target_rate, _, _ = qa.size(TargetPerf(target_itl=1e9, target_ttft=1e9))
        assert target_rate.rate_target_itl == pytest.approx(qa.rate_range.max)
        assert target_rate.rate_target_ttft == pytest.approx(qa.rate_range.max)

    def test_invalid_targets(self):
        qa = make_analyzer()
        with pytest.raises(AnalyzerError):
            qa.size(TargetPerf(target_itl=-1.0))


class TestEffectiveConcurrency:
    def test_identity(self):
        sp = ServiceParms(
            prefill=PrefillParms(gamma=5.0, delta=0.01),
            decode=DecodeParms(alpha=10.0, beta=1.0),
        )
        rs = RequestSize(avg_input_tokens=100, avg_output_tokens=10)
        n_true = 3.5
        serv_time = (5.0 + 0.01 * 100 * n_true) + (10.0 + 1.0 * n_true) * 9
        n = effective_concurrency(serv_time, sp, rs, max_batch_size=8)
        assert n == pytest.approx(n_true, rel=1e-9)

    def test_clamping(self):
        sp = ServiceParms(
            prefill=PrefillParms(gamma=5.0, delta=0.01),
            decode=DecodeParms(alpha=10.0, beta=1.0),
        )
        rs = RequestSize(avg_input_tokens=100, avg_output_tokens=10)
        assert effective_concurrency(0.0, sp, rs, 8) == 0.0
        assert effective_concurrency(1e9, sp, rs, 8) == 8.0

    def test_degenerate_denominator(self):
        sp = ServiceParms(
            prefill=PrefillParms(gamma=5.0, delta=0.0),
            decode=DecodeParms(alpha=10.0, beta=0.0),
        )
        rs = RequestSize(avg_input_tokens=0, avg_output_tokens=1)
        assert effective_concurrency(100.0, sp, rs, 8) == 8.0
        assert effective_concurrency(0.0, sp, rs, 8) == 0.0


class TestMG1:
    """M/G/1 extension (wva_amd/analyzer/mg1.py): P-K formula + SCV
    mapping.  The reference is Markovian-only; these check the exact
    special cases the approximation must reproduce."""

    def test_mm1_reduction(self):
        # scv=1 (exponential) must equal the textbook M/M/1 wait
        # Wq = rho / (mu - lam)
        lam, mu = 0.6, 1.0
        from wva_amd.analyzer import pollaczek_khinchine_wait

        w = pollaczek_khinchine_wait(lam, 1.0 / mu, scv=1.0)
        rho = lam / mu
        assert w == pytest.approx(rho / (mu - lam))

    def test_md1_half_wait(self):
        # deterministic service: exactly half the M/M/1 wait
        from wva_amd.analyzer import pollaczek_khinchine_wait

        w_exp = pollaczek_khinchine_wait(0.8, 1.0, scv=1.0)
        w_det = pollaczek_khinchine_wait(0.8, 1.0, scv=0.0)
        assert w_det == pytest.approx(w_exp / 2)

    def test_unstable_raises(self):
        from wva_amd.analyzer import pollaczek_khinchine_wait

        with pytest.raises(ValueError):
            pollaczek_khinchine_wait(1.0, 1.0, scv=1.0)
        with pytest.raises(ValueError):
            pollaczek_khinchine_wait(0.5, -1.0, scv=1.0)

    def test_zero_load(self):
        from wva_amd.analyzer import pollaczek_khinchine_wait

        assert pollaczek_khinchine_wait(0.0, 5.0, scv=0.5) == 0.0

    def test_scv_from_tokens_all_variable(self):
        # no fixed component: service SCV = token SCV
        from wva_amd.analyzer import service_scv_from_tokens

        assert service_scv_from_tokens(100.0, 0.7, 0.5, fixed_time=0.0) == pytest.approx(0.7)

    def test_scv_from_tokens_damped_by_fixed(self):
        # fixed part equal to the variable part: damping (1/2)^2
        from wva_amd.analyzer import service_scv_from_tokens

        scv = service_scv_from_tokens(100.0, 1.0, 0.5, fixed_time=50.0)
        assert scv == pytest.approx(0.25)

    def test_corrector_identity_and_scaling(self):
        from wva_amd.analyzer import MG1Corrector

        assert MG1Corrector(1.0).correct(10.0).wait == pytest.approx(10.0)
        m = MG1Corrector(0.0).correct(10.0)
        assert m.wait == pytest.approx(5.0)
        assert m.correction == pytest.approx(0.5)
        assert m.markovian_wait == 10.0
        with pytest.raises(ValueError):
            MG1Corrector(-0.1)
        with pytest.raises(ValueError):
            MG1Corrector(1.0).correct(float("nan"))

    def test_consistency_with_statedep_large_n(self):
        # sanity: for a single-slot exponential server (N=1, huge K) the
        # state-dependent chain's wait approaches the M/M/1 wait that
        # scv=1 P-K reproduces — ties the extension to the main analyzer
        from wva_amd.analyzer import pollaczek_khinchine_wait

        config = Configuration(
            max_batch_size=1,
            max_queue_size=2000,
            service_parms=ServiceParms(
                prefill=PrefillParms(gamma=0.0, delta=0.0),
                decode=DecodeParms(alpha=10.0, beta=0.0),
            ),
        )
        qa = QueueAnalyzer(config, RequestSize(avg_input_tokens=0, avg_output_tokens=2))
        serv_time_ms = 10.0  # one decode pass: (K-1) * alpha
        lam_per_s = 50.0  # rho = 0.5 (analyze takes req/s; services are ms)
        metrics = qa.analyze(lam_per_s)
        pk_ms = pollaczek_khinchine_wait(lam_per_s / 1000.0, serv_time_ms, scv=1.0)
        assert metrics.avg_wait_time == pytest.approx(pk_ms, rel=0.05)
