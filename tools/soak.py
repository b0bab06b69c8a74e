"""Closed-loop soak: drive the demo load staircase against a *scalable*
emulator fleet and record the autoscaler's decisions AND the observed
serving latency vs the SLO.

The in-process version of the reference demo's 8→16→24→16→8→0 req/s
staircase (docs/tutorials/demo.md), compressed in time, with the
actuation chain the reference only proves on hardware
(test/e2e-openshift/sharegpt_scaleup_test.go): each cycle the simulated
HPA resizes the emulator fleet to the recommended replica count, load is
re-split across the active instances, and the next stage's observed
TTFT/ITL (from the fleet's Prometheus histograms) is scored against the
class targets next to the analyzer's prediction (VERDICT r01 #2 —
attainment is measured, not predicted).

    python tools/soak.py --stage-seconds 8 --out soak.json
"""

from __future__ import annotations

import argparse
import json
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))
sys.path.insert(0, str(ROOT / "tools"))
sys.path.insert(0, str(ROOT / "tests"))

MODEL = "default/llama-8b"
# Premium class targets for llama-8b (tests/kube_fixtures.py)
TARGET_ITL_MS = 24.0
TARGET_TTFT_MS = 500.0
# emulator step-law parameters (mirrored into the VA profile)
ALPHA, BETA, GAMMA, DELTA = 12.0, 6.0, 4.0, 0.01
MAX_BATCH = 16


def run_soak(
    stages=(2.0, 4.0, 6.0, 4.0, 2.0, 0.0),
    stage_seconds: float = 8.0,
    max_replicas: int = 6,
    variants: int = 1,
    quiet: bool = False,
    scale_down_stabilization_s: float = 0.0,
) -> dict:
    """Run the closed loop; returns the result dict (importable from
    bench.py so BENCH records carry measured attainment)."""

    class _Args:
        pass

    args = _Args()
    args.stages = list(stages)
    args.stage_seconds = stage_seconds
    args.max_replicas = max_replicas
    args.variants = variants

    def emit(payload) -> None:
        if not quiet:
            print(json.dumps(payload))

    import logging
    import os

    os.environ.setdefault("WVA_RATE_WINDOW", "8s")

    import torch

    from wva_amd.controller.logger import log as wva_log

    wva_log.init(logging.ERROR)

    from prometheus_client import CollectorRegistry

    from vllm_emulator.engine import EmulatorSettings
    from vllm_emulator.fleet import EmulatorFleet
    from slo_observer import observe_latency, predict_latency, score
    from wva_amd.api import v1alpha1
    from wva_amd.controller import metrics as ctrl_metrics
    from wva_amd.controller.reconciler import VariantAutoscalingReconciler
    from wva_amd.promlib import PromlibAPI, Scraper, TimeSeriesStore
    from kube_fixtures import make_cluster, make_deployment, make_va
    from test_e2e import EmulatorProcess, drive_load, simulate_hpa

    registry = CollectorRegistry()
    ctrl_metrics.init_metrics(registry)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    cluster = make_cluster(opt_interval="1s")
    names, models = [], []
    for v in range(args.variants):
        name = "vllm-llama" if v == 0 else f"vllm-extra-{v}"
        model = MODEL if v == 0 else "default/llama-70b"
        names.append(name)
        models.append(model)
        make_deployment(cluster, name=name, replicas=1)
        make_va(cluster, name=name, model_id=model, max_batch=MAX_BATCH,
                alpha=str(ALPHA), beta=str(BETA), gamma=str(GAMMA), delta=str(DELTA))

    store = TimeSeriesStore()
    scraper = Scraper(store)
    prom = PromlibAPI(store)
    rec = VariantAutoscalingReconciler(
        cluster,
        prom,
        batched_analyzer=device == "cuda",
        analyzer_device=device if device == "cuda" else None,
    )

    settings = EmulatorSettings(
        model=MODEL,
        decode_alpha=ALPHA,
        decode_beta=BETA,
        prefill_gamma=GAMMA,
        prefill_delta=DELTA,
        avg_generated_len=25,
        tokens_distribution="deterministic",
        max_batch_size=MAX_BATCH,
        realtime=True,
    )

    import contextlib

    trajectory = []
    recommendations = []  # (wall_time, desired) for scale-down stabilization
    with contextlib.ExitStack() as stack:
        fleet = stack.enter_context(EmulatorFleet(settings, max_replicas=args.max_replicas))
        for url in fleet.urls:
            scraper.add_target(f"{url}/metrics", extra_labels={"namespace": "default"})
        extra_emus = []
        for v, model in enumerate(models[1:], start=1):
            emu = stack.enter_context(EmulatorProcess(settings.__class__(
                model=model, decode_alpha=ALPHA, decode_beta=BETA,
                prefill_gamma=GAMMA, prefill_delta=DELTA, avg_generated_len=25,
                tokens_distribution="deterministic", max_batch_size=MAX_BATCH,
                realtime=True)))
            scraper.add_target(f"{emu.base_url}/metrics", extra_labels={"namespace": "default"})
            extra_emus.append(emu)
        scraper.start(interval=0.5)
        try:
            for rate in args.stages:
                import threading

                threads = []
                for emu in extra_emus:
                    t = threading.Thread(
                        target=drive_load, args=(emu.base_url, 2.0, args.stage_seconds)
                    )
                    t.start()
                    threads.append(t)
                if rate > 0:
                    fleet.drive(rate, args.stage_seconds, MODEL)
                elif not threads:
                    time.sleep(max(args.stage_seconds, 10.0))
                for t in threads:
                    t.join()

                # observed latency over the stage that just ran, at the
                # fleet size chosen by the PREVIOUS cycle
                obs = observe_latency(prom, MODEL, "default")

                t0 = time.perf_counter()
                rec.reconcile()
                cycle_ms = (time.perf_counter() - t0) * 1000.0
                va = cluster.get(v1alpha1.VariantAutoscaling, "vllm-llama", "default")
                desired = va.status.desired_optimized_alloc.num_replicas
                for name in names:
                    simulate_hpa(cluster, registry, name=name)
                # actuate: the simulated HPA resizes the serving fleet.
                # With scale_down_stabilization_s > 0 it applies HPA's
                # scaleDown stabilizationWindowSeconds semantics — the
                # applied size is the MAX recommendation in the trailing
                # window (scale-up stays instant), mirroring the shipped
                # deploy/integrations/hpa.yaml (120 s)
                now_w = time.time()
                recommendations.append((now_w, desired))
                if scale_down_stabilization_s > 0:
                    cutoff = now_w - scale_down_stabilization_s
                    recommendations[:] = [
                        (t, d) for t, d in recommendations if t >= cutoff
                    ] or [(now_w, desired)]
                    applied = max(d for _, d in recommendations)
                else:
                    applied = desired
                fleet.scale(max(applied, 1))

                arrival_rpm = float(va.status.current_alloc.load.arrival_rate)
                per_replica = (arrival_rpm / 60.0) / max(desired, 1)
                pred = predict_latency(
                    ALPHA, BETA, GAMMA, DELTA, MAX_BATCH,
                    int(float(va.status.current_alloc.load.avg_input_tokens) or 32),
                    int(float(va.status.current_alloc.load.avg_output_tokens) or 25),
                    per_replica,
                )
                s = score(obs, pred, TARGET_TTFT_MS, TARGET_ITL_MS)
                entry = {
                    "offered_rps": rate,
                    "measured_arrival_rpm": arrival_rpm,
                    "desired_replicas": desired,
                    "fleet_replicas": fleet.replicas,
                    "cycle_ms": cycle_ms,
                    "metrics_ok": v1alpha1.is_condition_true(va, v1alpha1.TYPE_METRICS_AVAILABLE),
                    "optimized_ok": v1alpha1.is_condition_true(va, v1alpha1.TYPE_OPTIMIZATION_READY),
                    **s.as_dict(),
                }
                if args.variants > 1:
                    entry["extra_desired"] = [
                        cluster.get(v1alpha1.VariantAutoscaling, n, "default").status.desired_optimized_alloc.num_replicas
                        for n in names[1:]
                    ]
                trajectory.append(entry)
                emit(entry)
        finally:
            scraper.stop()

    result = {
        "device": device,
        "stage_seconds": args.stage_seconds,
        "max_replicas": args.max_replicas,
        "trajectory": trajectory,
    }
    # staircase sanity: rises with load, returns to baseline on idle
    peaks = [t["desired_replicas"] for t in trajectory]
    result["monotone_rise"] = peaks[2] >= peaks[1] >= peaks[0] >= 1
    result["returns_to_min"] = peaks[-1] == 1
    # observed attainment: loaded stages AFTER the fleet has been resized
    # at least once (stage 0 runs at the initial size)
    scored = [t for t in trajectory[1:] if t["offered_rps"] > 0]
    met = [t for t in scored if t["observed_met"]]
    result["observed_slo_attainment_pct"] = (
        100.0 * len(met) / len(scored) if scored else None
    )
    drifts = [t["itl_drift_pct"] for t in scored if t["itl_drift_pct"] is not None]
    result["mean_itl_drift_pct"] = sum(drifts) / len(drifts) if drifts else None
    emit({"summary": {
        k: result[k]
        for k in ("device", "monotone_rise", "returns_to_min",
                  "observed_slo_attainment_pct", "mean_itl_drift_pct")
    }})
    return result


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--stages", type=float, nargs="+", default=[2.0, 4.0, 6.0, 4.0, 2.0, 0.0])
    ap.add_argument("--stage-seconds", type=float, default=8.0)
    ap.add_argument("--max-replicas", type=int, default=6, help="fleet instances pre-started")
    ap.add_argument("--variants", type=int, default=1, help="number of variants (extras carry steady load)")
    ap.add_argument("--scale-down-stabilization", type=float, default=0.0,
                    help="HPA scaleDown stabilizationWindowSeconds to emulate (0 = apply instantly)")
    ap.add_argument("--out", default="")
    args = ap.parse_args()
    result = run_soak(
        stages=args.stages,
        stage_seconds=args.stage_seconds,
        max_replicas=args.max_replicas,
        variants=args.variants,
        scale_down_stabilization_s=args.scale_down_stabilization,
    )
    if args.out:
        with open(args.out, "w") as f:
            json.dump(result, f, indent=2)


if __name__ == "__main__":
    main()
