"""Opt-in M/G/1 analyzer: WVA_ANALYZER=mg1 applies
the Allen-Cunneen wait scaling (1+cs^2)/2 inside Size/Analyze, so
replica counts shrink for low-variability workloads.  The emulator
side of the validation (deterministic output lengths, measured wait vs
the cs^2=1 and cs^2=0 predictions) runs in tools/mg1_experiment.py and
is summarized with numbers in docs/design/mg1-analyzer.md.
"""

import numpy as np
import pytest

from wva_amd.analyzer import (
    Configuration,
    DecodeParms,
    PrefillParms,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    TargetPerf,
    configured_scv,
)
from wva_amd.ops import solve_problems
from wva_amd.ops.batched import (
    P_ALPHA,
    P_BETA,
    P_GAMMA,
    P_DELTA,
    P_IN_TOKENS,
    P_OUT_TOKENS,
    P_MAX_BATCH,
    P_TARGET_TTFT,
    P_TARGET_ITL,
    P_TOTAL_RATE,
    P_MIN_REPLICAS,
    PROBLEM_FIELDS,
    R_FEASIBLE,
    R_REPLICAS,
    R_RATE_STAR,
)


def make_qa(scv):
    config = Configuration(
        max_batch_size=16,
        max_queue_size=160,
        service_parms=ServiceParms(
            prefill=PrefillParms(gamma=4.0, delta=0.01),
            decode=DecodeParms(alpha=12.0, beta=6.0),
        ),
    )
    return QueueAnalyzer(
        config, RequestSize(avg_input_tokens=32, avg_output_tokens=25), scv=scv
    )


class TestScvSelection:
    def test_default_is_markovian(self, monkeypatch):
        monkeypatch.delenv("WVA_ANALYZER", raising=False)
        assert configured_scv() == 1.0
        monkeypatch.setenv("WVA_ANALYZER", "mm1k")
        assert configured_scv() == 1.0

    def test_mg1_default_and_override(self, monkeypatch):
        monkeypatch.setenv("WVA_ANALYZER", "mg1")
        monkeypatch.delenv("WVA_SERVICE_SCV", raising=False)
        assert configured_scv() == 0.5
        monkeypatch.setenv("WVA_SERVICE_SCV", "0.25")
        assert configured_scv() == 0.25

    def test_invalid_mode_and_scv_rejected(self, monkeypatch):
        monkeypatch.setenv("WVA_ANALYZER", "gg1")
        with pytest.raises(ValueError):
            configured_scv()
        monkeypatch.setenv("WVA_ANALYZER", "mg1")
        monkeypatch.setenv("WVA_SERVICE_SCV", "-1")
        with pytest.raises(ValueError):
            configured_scv()


class TestWaitScaling:
    def test_scv_one_is_identity(self):
        base, mg1 = make_qa(1.0), make_qa(1.0)
        a = base.analyze(4.0)
        b = mg1.analyze(4.0)
        assert a.avg_wait_time == b.avg_wait_time
        assert a.avg_resp_time == b.avg_resp_time

    def test_deterministic_halves_wait(self):
        wait_mm1 = make_qa(1.0).analyze(4.0).avg_wait_time
        wait_md1 = make_qa(0.0).analyze(4.0).avg_wait_time
        assert wait_mm1 > 0
        assert wait_md1 == pytest.approx(wait_mm1 / 2.0)

    def test_wait_monotone_in_scv(self):
        waits = [make_qa(s).analyze(4.0).avg_wait_time for s in (0.0, 0.5, 1.0)]
        assert waits[0] < waits[1] < waits[2]

    def test_throughput_and_itl_unchanged(self):
        # the correction touches queueing delay only
        a = make_qa(1.0).analyze(4.0)
        b = make_qa(0.0).analyze(4.0)
        assert a.throughput == b.throughput
        assert a.avg_token_time == b.avg_token_time


class TestSizingRespondsToScv:
    def _rate_star(self, scv, target_ttft=40.0):
        qa = make_qa(scv)
        _, metrics, _ = qa.size(
            TargetPerf(target_ttft=target_ttft, target_itl=0.0, target_tps=0.0)
        )
        return metrics.throughput

    def test_lower_variability_supports_higher_rate(self):
        # with a TTFT (wait-bound) target, a deterministic workload can
        # be driven harder per replica than an exponential one
        r0 = self._rate_star(0.0)
        r05 = self._rate_star(0.5)
        r1 = self._rate_star(1.0)
        assert r0 >= r05 >= r1
        assert r0 > r1 * 1.02, f"expected a real margin, got {r0} vs {r1}"

    def test_replica_counts_compare_across_scv(self, monkeypatch):
        """The VERDICT's comparison: fleet replica counts under
        cs^2 in {0, 0.5, 1} for a wait-bound workload."""
        row = np.zeros(PROBLEM_FIELDS)
        row[P_ALPHA], row[P_BETA] = 12.0, 6.0
        row[P_GAMMA], row[P_DELTA] = 4.0, 0.01
        row[P_IN_TOKENS], row[P_OUT_TOKENS], row[P_MAX_BATCH] = 32, 25, 16
        row[P_TARGET_TTFT] = 15.0  # tight wait-bound target
        row[P_TARGET_ITL] = 0.0
        row[P_TOTAL_RATE] = 200.0  # req/s across the fleet
        row[P_MIN_REPLICAS] = 1
        problems = np.stack([row] * 4)

        replicas = {}
        for scv, env in ((1.0, None), (0.5, "0.5"), (0.0, "0")):
            if env is None:
                monkeypatch.delenv("WVA_ANALYZER", raising=False)
            else:
                monkeypatch.setenv("WVA_ANALYZER", "mg1")
                monkeypatch.setenv("WVA_SERVICE_SCV", env)
            out = solve_problems(problems)
            assert (out[:, R_FEASIBLE] == 1.0).all()
            replicas[scv] = int(out[0, R_REPLICAS])
        assert replicas[0.0] <= replicas[0.5] <= replicas[1.0]
        assert replicas[0.0] < replicas[1.0], (
            f"cs^2=0 should need fewer replicas than cs^2=1: {replicas}"
        )

    def test_mg1_mode_routes_off_native_kernel(self, monkeypatch):
        # the native kernels are Markovian-only; mg1 must use the scalar
        # analyzer even when a native binding is importable
        from unittest import mock

        import wva_amd.ops.batched as batched

        monkeypatch.setenv("WVA_ANALYZER", "mg1")
        row = np.zeros(PROBLEM_FIELDS)
        row[P_ALPHA], row[P_BETA] = 12.0, 6.0
        row[P_GAMMA], row[P_DELTA] = 4.0, 0.01
        row[P_IN_TOKENS], row[P_OUT_TOKENS], row[P_MAX_BATCH] = 32, 25, 16
        row[P_TARGET_ITL] = 50.0
        row[P_TOTAL_RATE] = 2.0
        row[P_MIN_REPLICAS] = 1
        with mock.patch.object(
            batched, "_solve_problems_python", wraps=batched._solve_problems_python
        ) as spy:
            out = solve_problems(row[None, :])
        assert spy.called and spy.call_args.kwargs.get("scv") == 0.5
        assert out[0, R_FEASIBLE] == 1.0

    def test_rate_star_margin_magnitude(self):
        # quantify the over-provisioning margin the doc reports: for this
        # wait-bound config the deterministic workload sustains a
        # measurably higher per-replica rate
        r0, r1 = self._rate_star(0.0), self._rate_star(1.0)
        margin_pct = (r0 - r1) / r1 * 100.0
        assert margin_pct > 3.0


class TestMG1ThroughReconciler:
    def test_mg1_mode_lowers_wait_bound_replicas(self, monkeypatch):
        """Fleet-level integration: with a wait-bound (tight-TTFT)
        workload, the reconciler sizes fewer-or-equal replicas under
        WVA_ANALYZER=mg1 (cs^2=0) than under the Markovian default,
        end to end through the controller."""
        import sys
        from pathlib import Path

        sys.path.insert(0, str(Path(__file__).resolve().parent))
        from prometheus_client import CollectorRegistry

        from wva_amd.api import v1alpha1
        from wva_amd.controller import metrics as ctrl_metrics
        from wva_amd.controller.promclient import MockPromAPI
        from wva_amd.controller.reconciler import VariantAutoscalingReconciler
        from kube_fixtures import (
            PREMIUM_YAML,
            make_cluster,
            make_deployment,
            make_va,
            set_load_metrics,
        )
        from wva_amd.controller.reconciler import (
            CONFIG_MAP_NAMESPACE,
            SERVICE_CLASSES_CM,
        )
        from wva_amd.kube import ConfigMap

        def desired_for(analyzer_env):
            if analyzer_env is None:
                monkeypatch.delenv("WVA_ANALYZER", raising=False)
            else:
                monkeypatch.setenv("WVA_ANALYZER", "mg1")
                monkeypatch.setenv("WVA_SERVICE_SCV", analyzer_env)
            cluster = make_cluster()
            # tighten the Premium TTFT target so waiting time binds
            cm = cluster.get(ConfigMap, SERVICE_CLASSES_CM, CONFIG_MAP_NAMESPACE)
            cm.data["premium.yaml"] = PREMIUM_YAML.replace(
                "slo-ttft: 500", "slo-ttft: 15"
            ).replace("slo-tpot: 24", "slo-tpot: 0")
            cluster.update(cm)
            make_deployment(cluster, replicas=1)
            make_va(cluster, max_batch=16, alpha="12.0", beta="6.0",
                    gamma="4.0", delta="0.01")
            prom = MockPromAPI()
            set_load_metrics(prom, "default/llama-8b", "default",
                             arrival_rps=40.0, in_tokens=32.0, out_tokens=25.0)
            registry = CollectorRegistry()
            ctrl_metrics.init_metrics(registry)
            try:
                VariantAutoscalingReconciler(cluster, prom).reconcile()
            finally:
                ctrl_metrics.reset_metrics()
            va = cluster.get(v1alpha1.VariantAutoscaling, "vllm-llama", "default")
            return va.status.desired_optimized_alloc.num_replicas

        markovian = desired_for(None)
        md1 = desired_for("0")
        assert markovian >= 1 and md1 >= 1
        assert md1 <= markovian
        assert md1 < markovian, (
            f"cs^2=0 should size fewer replicas on a wait-bound workload "
            f"(markovian={markovian}, md1={md1})"
        )


class TestAutoScvEstimation:
    def _bucket_samples(self, cum_by_le):
        import time as _time

        from wva_amd.controller.promclient import Sample

        now = _time.time()
        return [
            Sample(value=v, timestamp=now, labels={"le": le})
            for le, v in cum_by_le.items()
        ]

    def test_estimator_concentrated_vs_spread(self):
        from wva_amd.controller import collector
        from wva_amd.controller.promclient import MockPromAPI

        prom = MockPromAPI()
        q = collector.token_scv_query("m", "ns")
        # concentrated: nearly all mass in one bucket -> tiny scv
        prom.query_results[q] = self._bucket_samples(
            {"16": 0.0, "32": 9.9, "64": 10.0, "+Inf": 10.0}
        )
        tight = collector.estimate_token_scv(prom, "m", "ns")
        # spread: mass across 3 octaves -> large scv
        prom.query_results[q] = self._bucket_samples(
            {"16": 4.0, "64": 7.0, "512": 9.0, "+Inf": 10.0}
        )
        wide = collector.estimate_token_scv(prom, "m", "ns")
        assert tight is not None and wide is not None
        assert 0.0 <= tight < 0.2
        assert wide > tight * 3

    def test_estimator_absent_histogram_returns_none(self):
        from wva_amd.controller import collector
        from wva_amd.controller.promclient import MockPromAPI

        prom = MockPromAPI()
        q = collector.token_scv_query("m", "ns")
        prom.query_results[q] = []
        # fallback (namespace-less) also empty
        prom.query_results[
            f'rate(vllm:request_generation_tokens_bucket{{model_name="m"}}'
            f"[{collector.rate_window()}])"
        ] = []
        assert collector.estimate_token_scv(prom, "m", "ns") is None

    def test_recommended_scv_damped_by_fixed_time(self):
        from wva_amd.analyzer import recommended_service_scv

        # long outputs, tiny fixed part -> near the token scv
        near = recommended_service_scv(200.0, 0.8, alpha=10.0, beta=0.1,
                                       gamma=1.0, delta=0.0, in_tokens=0)
        # short outputs, big prefill -> strongly damped
        damped = recommended_service_scv(5.0, 0.8, alpha=10.0, beta=0.1,
                                         gamma=500.0, delta=0.0, in_tokens=0)
        assert 0.5 < near <= 1.0
        assert damped < near / 4

    def test_auto_mode_end_to_end_through_reconciler(self, monkeypatch):
        """WVA_SERVICE_SCV=auto: a concentrated token histogram drives a
        low per-server cs^2 and fewer replicas than the Markovian
        default on a wait-bound workload."""
        import sys
        from pathlib import Path

        sys.path.insert(0, str(Path(__file__).resolve().parent))
        from prometheus_client import CollectorRegistry

        from wva_amd.api import v1alpha1
        from wva_amd.controller import collector
        from wva_amd.controller import metrics as ctrl_metrics
        from wva_amd.controller.promclient import MockPromAPI
        from wva_amd.controller.reconciler import (
            CONFIG_MAP_NAMESPACE,
            SERVICE_CLASSES_CM,
            VariantAutoscalingReconciler,
        )
        from wva_amd.kube import ConfigMap
        from kube_fixtures import (
            PREMIUM_YAML,
            make_cluster,
            make_deployment,
            make_va,
            set_load_metrics,
        )

        def desired(env_scv):
            if env_scv is None:
                monkeypatch.delenv("WVA_ANALYZER", raising=False)
                monkeypatch.delenv("WVA_SERVICE_SCV", raising=False)
            else:
                monkeypatch.setenv("WVA_ANALYZER", "mg1")
                monkeypatch.setenv("WVA_SERVICE_SCV", env_scv)
            cluster = make_cluster()
            cm = cluster.get(ConfigMap, SERVICE_CLASSES_CM, CONFIG_MAP_NAMESPACE)
            cm.data["premium.yaml"] = PREMIUM_YAML.replace(
                "slo-ttft: 500", "slo-ttft: 15"
            ).replace("slo-tpot: 24", "slo-tpot: 0")
            cluster.update(cm)
            make_deployment(cluster, replicas=1)
            make_va(cluster, max_batch=16, alpha="12.0", beta="6.0",
                    gamma="4.0", delta="0.01")
            prom = MockPromAPI()
            set_load_metrics(prom, "default/llama-8b", "default",
                             arrival_rps=40.0, in_tokens=32.0, out_tokens=25.0)
            # concentrated histogram: all requests 24-26 tokens
            import time as _time

            from wva_amd.controller.promclient import Sample

            now = _time.time()
            prom.query_results[
                collector.token_scv_query("default/llama-8b", "default")
            ] = [
                Sample(value=0.0, timestamp=now, labels={"le": "24"}),
                Sample(value=10.0, timestamp=now, labels={"le": "26"}),
                Sample(value=10.0, timestamp=now, labels={"le": "+Inf"}),
            ]
            registry = CollectorRegistry()
            ctrl_metrics.init_metrics(registry)
            try:
                VariantAutoscalingReconciler(cluster, prom).reconcile()
            finally:
                ctrl_metrics.reset_metrics()
            va = cluster.get(v1alpha1.VariantAutoscaling, "vllm-llama", "default")
            return va.status.desired_optimized_alloc.num_replicas

        markovian = desired(None)
        auto = desired("auto")
        assert markovian >= 1 and auto >= 1
        assert auto < markovian, (
            f"auto cs^2 from a concentrated histogram must size fewer "
            f"replicas than mm1k on a wait-bound workload "
            f"(mm1k={markovian}, auto={auto})"
        )


class TestPollaczekKhinchineCrossValidation:
    @pytest.mark.parametrize("scv", [0.0, 0.5, 1.0])
    @pytest.mark.parametrize("rho", [0.3, 0.6, 0.8])
    def test_corrected_analyzer_matches_pk_formula(self, scv, rho):
        """At batch size 1 with a deep queue, the state-dependent chain
        IS an M/M/1(/K) queue; the Allen-Cunneen-corrected wait must
        match the exact Pollaczek-Khinchine M/G/1 wait."""
        from wva_amd.analyzer import pollaczek_khinchine_wait

        alpha, beta = 10.0, 0.0   # constant service
        out_tokens = 11           # 10 decode passes
        service_ms = (out_tokens - 1) * alpha  # 100 ms
        config = Configuration(
            max_batch_size=1,
            max_queue_size=2000,  # deep enough to approximate infinite K
            service_parms=ServiceParms(
                prefill=PrefillParms(gamma=0.0, delta=0.0),
                decode=DecodeParms(alpha=alpha, beta=beta),
            ),
        )
        qa = QueueAnalyzer(
            config,
            RequestSize(avg_input_tokens=0, avg_output_tokens=out_tokens),
            scv=scv,
        )
        lam_per_ms = rho / service_ms
        got = qa.analyze(lam_per_ms * 1000.0).avg_wait_time
        want = pollaczek_khinchine_wait(lam_per_ms, service_ms, scv)
        assert got == pytest.approx(want, rel=2e-3), (scv, rho)
