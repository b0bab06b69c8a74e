"""Hardware e2e (MI355X): the full autoscaling chain with the analyze
phase running through the gfx950 kernel.

Analog of the reference's OpenShift hardware e2e tier (SURVEY.md §4.4:
live load -> WVA recommends scale-up -> external metrics -> HPA raises ->
Deployment follows), executed on a real MI355X box via gpurun."""

import time

import pytest
from prometheus_client import CollectorRegistry

pytestmark = pytest.mark.gpu


@pytest.fixture()
def registry():
    from wva_amd.controller import metrics as ctrl_metrics

    reg = CollectorRegistry()
    ctrl_metrics.init_metrics(reg)
    yield reg
    ctrl_metrics.reset_metrics()


def test_gpu_analyzer_scaleup_chain(registry, monkeypatch):
    import torch

    assert torch.cuda.is_available()
    monkeypatch.setenv("WVA_RATE_WINDOW", "8s")

    from vllm_emulator.engine import EmulatorSettings
    from wva_amd.api import v1alpha1
    from wva_amd.controller.reconciler import VariantAutoscalingReconciler
    from wva_amd.kube import Deployment
    from wva_amd.promlib import PromlibAPI, Scraper, TimeSeriesStore
    from kube_fixtures import make_cluster, make_deployment, make_va
    from test_e2e import EmulatorProcess, drive_load, simulate_hpa

    settings = EmulatorSettings(
        model="default/llama-8b",
        decode_alpha=12.0,
        decode_beta=6.0,
        prefill_gamma=4.0,
        prefill_delta=0.01,
        avg_generated_len=25,
        tokens_distribution="deterministic",
        max_batch_size=16,
        realtime=True,
    )
    cluster = make_cluster(opt_interval="1s")
    make_deployment(cluster, replicas=1)
    make_va(cluster, max_batch=16, alpha="12.0", beta="6.0", gamma="4.0", delta="0.01")

    store = TimeSeriesStore()
    scraper = Scraper(store)
    prom = PromlibAPI(store)

    from vllm_emulator.fleet import EmulatorFleet
    from slo_observer import observe_latency

    with EmulatorFleet(settings, max_replicas=4) as fleet:
        for url in fleet.urls:
            scraper.add_target(f"{url}/metrics", extra_labels={"namespace": "default"})
        scraper.start(interval=0.5)
        try:
            # the analyze phase runs on the GPU (fails loudly if the native
            # extension is missing on this box)
            rec = VariantAutoscalingReconciler(
                cluster, prom, batched_analyzer=True, analyzer_device="cuda"
            )
            time.sleep(2.0)
            rec.reconcile()
            fleet.drive(5.0, 6.0, "default/llama-8b")
            # under-provisioned: one instance serves 5 rps against a
            # binding ITL SLO — the observed latency breaches the target
            obs_before = observe_latency(prom, "default/llama-8b", "default")
            rec.reconcile()
            va = cluster.get(v1alpha1.VariantAutoscaling, "vllm-llama", "default")
            desired = va.status.desired_optimized_alloc.num_replicas
            assert desired >= 2  # WVA recommends scale-up
            assert v1alpha1.is_condition_true(va, v1alpha1.TYPE_OPTIMIZATION_READY)
            # external autoscaler follows the gauge; Deployment resizes
            assert simulate_hpa(cluster, registry) == desired
            assert cluster.get(Deployment, "vllm-llama", "default").spec.replicas == desired

            # actuation closes the loop: after the fleet grows to the
            # recommendation, OBSERVED serving latency returns under the
            # SLO (the reference's hardware-e2e relation,
            # sharegpt_scaleup_test.go:39-253) — measured, not predicted
            fleet.scale(desired)
            fleet.drive(5.0, 6.0, "default/llama-8b")
            obs_after = observe_latency(prom, "default/llama-8b", "default")
            target_itl = 24.0  # Premium slo-tpot (kube_fixtures.py)
            assert obs_after.itl_ms <= target_itl * 1.1, (
                f"scaled fleet still over ITL target: {obs_after.itl_ms:.1f}ms"
            )
            assert obs_after.itl_ms <= obs_before.itl_ms + 1.0, (
                "scaling should not worsen observed ITL"
            )
        finally:
            scraper.stop()
