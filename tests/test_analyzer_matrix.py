"""Port of the reference's analyzer boundary-precision matrix
(/root/reference/pkg/analyzer/utils_test.go, 644 LoC — VERDICT r01 #5):
tolerance-comparison table, binary-search table incl. error propagation
and degenerate ranges, edge cases (constant / step / zero-range), the
analyzer-function integration searches, and the precision gate.  Test
names trace to the Go functions they port.
"""

import pytest

from wva_amd.analyzer import (
    Configuration,
    DecodeParms,
    PrefillParms,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
)
from wva_amd.analyzer.search import (
    AboveRegion,
    BelowRegion,
    InRegion,
    binary_search,
    within_tolerance,
)


class TestWithinToleranceTable:
    # utils_test.go:9 TestWithinTolerance
    @pytest.mark.parametrize(
        "name,x,value,tolerance,expected",
        [
            ("exact match", 1.0, 1.0, 0.01, True),
            ("within tolerance", 1.005, 1.0, 0.01, True),
            ("outside tolerance", 1.02, 1.0, 0.01, False),
            ("zero value", 0.1, 0.0, 0.01, False),
            # exact match wins regardless of a negative tolerance
            ("negative tolerance", 1.0, 1.0, -0.01, True),
            ("both zero", 0.0, 0.0, 0.01, True),
        ],
    )
    def test_table(self, name, x, value, tolerance, expected):
        assert within_tolerance(x, value, tolerance) is expected


def quadratic(x):
    return x * x


def linear(x):
    return 2 * x


def negative_linear(x):
    return -x


class EvalBoom(Exception):
    pass


def error_func(x):
    if x > 5.0:
        raise EvalBoom("x too large")
    return x


class TestBinarySearchTable:
    # utils_test.go:72 TestBinarySearch
    @pytest.mark.parametrize(
        "name,x_min,x_max,y_target,fn,expected_ind,tol",
        [
            ("find square root", 0.0, 10.0, 4.0, quadratic, InRegion, 0.1),
            ("linear target in range", 1.0, 5.0, 6.0, linear, InRegion, 0.1),
            ("linear target below range", 2.0, 5.0, 1.0, linear, BelowRegion, 0.1),
            ("linear target above range", 1.0, 3.0, 10.0, linear, AboveRegion, 0.1),
            ("decreasing target in range", 1.0, 5.0, -3.0, negative_linear, InRegion, 0.1),
            ("target at boundary", 1.0, 5.0, 2.0, linear, InRegion, 0.1),
        ],
    )
    def test_table(self, name, x_min, x_max, y_target, fn, expected_ind, tol):
        x_star, ind = binary_search(x_min, x_max, y_target, fn)
        assert ind == expected_ind
        if ind == InRegion:
            assert abs(fn(x_star) - y_target) <= tol
        if ind == BelowRegion:
            assert x_star == x_min
        if ind == AboveRegion:
            assert x_star == x_max

    def test_invalid_range_raises(self):
        # Go returns err for xMin > xMax; here that is a ValueError
        with pytest.raises(ValueError):
            binary_search(5.0, 1.0, 3.0, linear)

    def test_function_evaluation_error_propagates(self):
        # Go's evalFunc error return; our eval functions raise
        with pytest.raises(EvalBoom):
            binary_search(4.0, 6.0, 5.0, error_func)


class TestBinarySearchEdgeCases:
    # utils_test.go:225 TestBinarySearch_EdgeCases

    def test_constant_function_target_matches(self):
        x_star, ind = binary_search(1.0, 10.0, 5.0, lambda x: 5.0)
        assert ind == InRegion  # boundary evaluation hits the target

    def test_constant_function_target_doesnt_match(self):
        # flat curve: classified by value only, never by noise-direction
        x_star, ind = binary_search(1.0, 10.0, 3.0, lambda x: 5.0)
        assert ind == BelowRegion and x_star == 1.0
        x_star, ind = binary_search(1.0, 10.0, 8.0, lambda x: 5.0)
        assert ind == AboveRegion and x_star == 10.0

    def test_step_function(self):
        step = lambda x: 1.0 if x < 3.0 else 10.0
        x_star, ind = binary_search(1.0, 5.0, 5.0, step)
        # target sits inside the step's jump; search converges to the
        # discontinuity without diverging or erroring
        assert ind == InRegion
        assert 2.9 <= x_star <= 3.1

    def test_zero_range(self):
        x_star, ind = binary_search(3.0, 3.0, 6.0, lambda x: 2 * x)
        assert x_star == 3.0 and ind == InRegion


class TestBinarySearchWithAnalyzerFunctions:
    # utils_test.go:521 TestBinarySearchWithAnalyzerFunctions — the same
    # integration, through QueueAnalyzer's bound eval closures (the
    # package-global eval state of the Go version was deliberately
    # eliminated)
    @pytest.fixture()
    def qa(self):
        config = Configuration(
            max_batch_size=4,
            max_queue_size=8,
            service_parms=ServiceParms(
                prefill=PrefillParms(gamma=10.0, delta=0.001),
                decode=DecodeParms(alpha=1.0, beta=0.01),
            ),
        )
        return QueueAnalyzer(
            config, RequestSize(avg_input_tokens=100, avg_output_tokens=10)
        )

    @pytest.mark.parametrize(
        "target,evalname",
        [
            (25.0, "_eval_ttft"),   # msec target TTFT
            (2.0, "_eval_itl"),     # msec target inter-token latency
        ],
    )
    def test_search_with_eval_functions(self, qa, target, evalname):
        lam_min = qa.rate_range.min / 1000.0
        lam_max = qa.rate_range.max / 1000.0
        fn = getattr(qa, evalname)
        x_star, ind = binary_search(lam_min, lam_max, target, fn)
        assert lam_min <= x_star <= lam_max
        if ind == InRegion:
            assert abs(fn(x_star) - target) <= max(0.1, 1e-6 * target)

    def test_search_result_is_monotone_consistent(self, qa):
        # a tighter ITL target must not allow a higher lambda
        lam_min = qa.rate_range.min / 1000.0
        lam_max = qa.rate_range.max / 1000.0
        lam_loose, _ = binary_search(lam_min, lam_max, 2.0, qa._eval_itl)
        lam_tight, _ = binary_search(lam_min, lam_max, 1.1, qa._eval_itl)
        assert lam_tight <= lam_loose + 1e-12


class TestBinarySearchPrecision:
    # utils_test.go:610 TestBinarySearchPrecision
    def test_linear_precision(self):
        fn = lambda x: 2 * x + 3
        x_star, ind = binary_search(1.0, 5.0, 9.0, fn)
        assert ind == InRegion
        assert abs(x_star - 3.0) <= 1e-3
        assert abs(fn(x_star) - 9.0) <= 1e-3

    def test_precision_tracks_tolerance(self):
        # the float64 port must be at least as tight as the float32
        # reference: relative tolerance 1e-6 at the y level
        fn = lambda x: 2 * x + 3
        x_star, ind = binary_search(1.0, 5.0, 9.0, fn)
        assert ind == InRegion
        assert abs(fn(x_star) - 9.0) / 9.0 <= 1e-6

    def test_boundary_classification_is_exact_at_edges(self):
        # targets epsilon outside the reachable band must classify as
        # below/above, never as a sloppy in-region hit
        fn = lambda x: 2 * x  # range [2, 10] over x in [1, 5]
        eps = 1e-3
        _, ind = binary_search(1.0, 5.0, 2.0 - eps, fn)
        assert ind == BelowRegion
        _, ind = binary_search(1.0, 5.0, 10.0 + eps, fn)
        assert ind == AboveRegion
        # exactly representable edge values hit in-region via the
        # boundary evaluations
        _, ind = binary_search(1.0, 5.0, 2.0, fn)
        assert ind == InRegion
        _, ind = binary_search(1.0, 5.0, 10.0, fn)
        assert ind == InRegion


def _valid_parms():
    return ServiceParms(
        prefill=PrefillParms(gamma=10.0, delta=0.001),
        decode=DecodeParms(alpha=1.0, beta=0.01),
    )


class TestConfigurationCheckTable:
    # queueanalyzer_test.go:92 TestConfiguration_Check
    @pytest.mark.parametrize(
        "name,max_batch,max_queue,parms,want_err",
        [
            ("valid configuration", 8, 16, "valid", False),
            ("zero max batch size", 0, 16, "valid", True),
            ("negative max batch size", -1, 16, "valid", True),
            ("negative max queue size", 8, -1, "valid", True),
            ("nil service parameters", 8, 16, None, True),
        ],
    )
    def test_table(self, name, max_batch, max_queue, parms, want_err):
        from wva_amd.analyzer import AnalyzerError

        config = Configuration(
            max_batch_size=max_batch,
            max_queue_size=max_queue,
            service_parms=_valid_parms() if parms == "valid" else None,
        )
        rs = RequestSize(avg_input_tokens=100, avg_output_tokens=10)
        if want_err:
            with pytest.raises(AnalyzerError):
                QueueAnalyzer(config, rs)
        else:
            assert QueueAnalyzer(config, rs) is not None


class TestNewQueueAnalyzerTable:
    # queueanalyzer_test.go:26 TestNewQueueAnalyzer — constructor accepts
    # every degenerate-but-valid request shape and rejects the invalid ones
    @pytest.mark.parametrize(
        "name,in_tok,out_tok,want_err",
        [
            ("no prefill", 0, 10, False),
            ("no prefill, one output token", 0, 1, False),
            ("no decode", 100, 1, False),
            ("mixed prefill and decode", 200, 20, False),
            ("zero input and output tokens", 0, 0, True),
            ("negative tokens", -1, -1, True),
            ("no decode, no first output token", 50, 0, True),
        ],
    )
    def test_table(self, name, in_tok, out_tok, want_err):
        from wva_amd.analyzer import AnalyzerError

        config = Configuration(
            max_batch_size=8, max_queue_size=16, service_parms=_valid_parms()
        )
        rs = RequestSize(avg_input_tokens=in_tok, avg_output_tokens=out_tok)
        if want_err:
            with pytest.raises(AnalyzerError):
                QueueAnalyzer(config, rs)
        else:
            qa = QueueAnalyzer(config, rs)
            # a constructible analyzer must expose a usable rate range
            assert qa.rate_range.max > qa.rate_range.min > 0


class TestRequestSizeCheckTable:
    # queueanalyzer_test.go:178 TestRequestSize_Check
    @pytest.mark.parametrize(
        "in_tok,out_tok,want_err",
        [
            (100, 10, False),
            (0, 10, False),      # zero input = decode-only, valid
            (-1, 10, True),
            (100, 0, True),      # must produce at least one token
            (100, -5, True),
        ],
    )
    def test_table(self, in_tok, out_tok, want_err):
        from wva_amd.analyzer import AnalyzerError

        config = Configuration(
            max_batch_size=8, max_queue_size=16, service_parms=_valid_parms()
        )
        if want_err:
            with pytest.raises(AnalyzerError):
                QueueAnalyzer(
                    config,
                    RequestSize(avg_input_tokens=in_tok, avg_output_tokens=out_tok),
                )
        else:
            QueueAnalyzer(
                config, RequestSize(avg_input_tokens=in_tok, avg_output_tokens=out_tok)
            )


class TestPrefillTimeTable:
    # queueanalyzer_test.go:226 TestPrefillParms_PrefillTime
    @pytest.mark.parametrize(
        "name,in_tok,batch,expected",
        [
            ("no input tokens", 0, 4.0, 0.0),
            ("small batch", 1000, 1.0, 11.0),
            ("large batch", 2000, 8.0, 26.0),
            ("fractional batch size", 500, 2.5, 11.25),
        ],
    )
    def test_table(self, name, in_tok, batch, expected):
        prefill = PrefillParms(gamma=10.0, delta=0.001)
        assert prefill.prefill_time(in_tok, batch) == pytest.approx(expected)


class TestDecodeTimeTable:
    # queueanalyzer_test.go:274 TestDecodeParms_DecodeTime
    @pytest.mark.parametrize(
        "batch,expected",
        [
            (1.0, 1.01),
            (8.0, 1.08),
            (0.0, 1.0),
            (2.5, 1.025),
        ],
    )
    def test_table(self, batch, expected):
        decode = DecodeParms(alpha=1.0, beta=0.01)
        assert decode.decode_time(batch) == pytest.approx(expected)


class TestAnalyzeRateSweepTable:
    # queueanalyzer_test.go:357 TestQueueAnalyzer_Analyze — rate sweep
    # across the operating range; every successful point returns sane
    # metrics, zero/negative/beyond-max rates raise
    def _qa(self):
        config = Configuration(
            max_batch_size=8, max_queue_size=16, service_parms=_valid_parms()
        )
        return QueueAnalyzer(
            config, RequestSize(avg_input_tokens=100, avg_output_tokens=10)
        )

    @pytest.mark.parametrize(
        "name,rate_of_range,want_err",
        [
            ("zero request rate", lambda r: 0.0, True),
            ("negative request rate", lambda r: -1.0, True),
            ("low request rate", lambda r: r.min * 0.5, False),
            ("medium request rate", lambda r: (r.min + r.max) * 0.5, False),
            ("high request rate within bounds", lambda r: r.max * 0.9, False),
            ("request rate exceeding maximum", lambda r: r.max * 1.1, True),
        ],
    )
    def test_table(self, name, rate_of_range, want_err):
        from wva_amd.analyzer import AnalyzerError

        qa = self._qa()
        rate = rate_of_range(qa.rate_range)
        if want_err:
            with pytest.raises(AnalyzerError):
                qa.analyze(rate)
            return
        m = qa.analyze(rate)
        assert m.throughput >= 0
        assert m.avg_resp_time >= 0
        assert m.avg_wait_time >= 0
        assert m.avg_num_in_serv >= 0
        assert 0.0 <= m.rho <= 1.0
        assert m.avg_prefill_time >= 0
        assert m.avg_token_time >= 0

    def test_metrics_monotone_in_rate(self):
        # queueing sanity across the sweep: wait and occupancy grow with λ
        qa = self._qa()
        rates = [qa.rate_range.min * 0.5,
                 (qa.rate_range.min + qa.rate_range.max) * 0.5,
                 qa.rate_range.max * 0.9]
        ms = [qa.analyze(r) for r in rates]
        assert ms[0].avg_wait_time <= ms[1].avg_wait_time <= ms[2].avg_wait_time
        assert ms[0].avg_num_in_serv <= ms[1].avg_num_in_serv <= ms[2].avg_num_in_serv
        assert ms[0].throughput <= ms[1].throughput <= ms[2].throughput


class TestDiagnosticStrings:
    # queueanalyzer_test.go:602 TestStringMethods — every analyzer type
    # renders a non-empty diagnostic string naming its fields (dataclass
    # reprs here; the Go side hand-writes String())
    def test_reprs_name_fields(self):
        parms = _valid_parms()
        config = Configuration(
            max_batch_size=8, max_queue_size=16, service_parms=parms
        )
        assert "max_batch_size" in repr(config)
        assert "gamma" in repr(parms.prefill) and "delta" in repr(parms.prefill)
        assert "alpha" in repr(parms.decode) and "beta" in repr(parms.decode)
        rs = RequestSize(avg_input_tokens=100, avg_output_tokens=10)
        assert "avg_input_tokens" in repr(rs)
        qa = QueueAnalyzer(config, rs)
        assert "min" in repr(qa.rate_range) and "max" in repr(qa.rate_range)
        m = qa.analyze(qa.rate_range.max * 0.5)
        assert "throughput" in repr(m) and "rho" in repr(m)


class TestMM1KUtilizationGrid:
    # queuemodel_test.go:152 TestMM1KModel_ProbabilityCalculation — the
    # utilization grid incl. the lambda == mu boundary
    @pytest.mark.parametrize(
        "name,lam,mu",
        [
            ("low utilization", 0.5, 2.0),
            ("medium utilization", 1.5, 2.0),
            ("high utilization", 1.9, 2.0),
            ("equal arrival and service rates", 2.0, 2.0),
        ],
    )
    def test_probability_grid(self, name, lam, mu):
        from wva_amd.analyzer import MM1KModel

        model = MM1KModel(3)
        model.solve(lam, mu)
        if not model.is_valid:
            pytest.skip("invalid model for this point")
        assert (model.p >= 0).all()
        assert float(model.p.sum()) == pytest.approx(1.0, abs=1e-9)
        assert 0.0 <= model.throughput <= lam


class TestMM1KvsStateDependentComparison:
    # queuemodel_test.go:461 TestMM1Models_Comparison: constant service
    # rates must make the state-dependent chain coincide with M/M/1/K
    def test_constant_rates_coincide(self):
        import numpy as np

        from wva_amd.analyzer import MM1KModel, MM1ModelStateDependent

        K, rate, lam = 5, 3.0, 1.5
        mm1k = MM1KModel(K)
        mm1k.solve(lam, rate)
        statedep = MM1ModelStateDependent(K, np.full(K, rate))
        statedep.solve(lam, 1.0)
        assert mm1k.is_valid and statedep.is_valid
        assert statedep.avg_num_in_system == pytest.approx(
            mm1k.avg_num_in_system, abs=1e-9
        )
        assert statedep.throughput == pytest.approx(mm1k.throughput, abs=1e-9)
        assert statedep.avg_resp_time == pytest.approx(mm1k.avg_resp_time, abs=1e-9)
