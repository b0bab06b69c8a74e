"""Closed-loop soak: drive the demo load staircase against the emulator
and record the autoscaler's decision trajectory.

The in-process version of the reference demo's 8→16→24→16→8→0 req/s
staircase (docs/tutorials/demo.md), compressed in time.  On a GPU box the
analyze phase runs through the gfx950 kernel.

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


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--stages", type=float, nargs="+", default=[2.0, 4.0, 6.0, 4.0, 2.0, 0.0])
    ap.add_argument("--stage-seconds", type=float, default=8.0)
    ap.add_argument("--variants", type=int, default=1, help="number of variants/emulators")
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    import logging
    import os

    os.environ.setdefault("WVA_RATE_WINDOW", "8s")

    import torch

    from wva_amd.controller.logger import log as wva_log

    wva_log.init(logging.ERROR)

    from prometheus_client import CollectorRegistry

    from vllm_emulator.engine import EmulatorSettings
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
    # variant 0 runs the staircase; additional variants carry steady load
    names, models = [], []
    for v in range(args.variants):
        name = "vllm-llama" if v == 0 else f"vllm-extra-{v}"
        model = "default/llama-8b" if v == 0 else f"default/llama-70b"
        names.append(name)
        models.append(model)
        make_deployment(cluster, name=name, replicas=1)
        make_va(cluster, name=name, model_id=model, max_batch=16,
                alpha="12.0", beta="6.0", gamma="4.0", delta="0.01")

    store = TimeSeriesStore()
    scraper = Scraper(store)
    rec = VariantAutoscalingReconciler(
        cluster,
        PromlibAPI(store),
        batched_analyzer=device == "cuda",
        analyzer_device=device if device == "cuda" else None,
    )

    import contextlib

    trajectory = []
    with contextlib.ExitStack() as stack:
        emus = []
        for v, model in enumerate(models):
            settings = EmulatorSettings(
                model=model,
                decode_alpha=12.0,
                decode_beta=6.0,
                prefill_gamma=4.0,
                prefill_delta=0.01,
                avg_generated_len=25,
                tokens_distribution="deterministic",
                max_batch_size=16,
                realtime=True,
            )
            emu = stack.enter_context(EmulatorProcess(settings))
            scraper.add_target(f"{emu.base_url}/metrics", extra_labels={"namespace": "default"})
            emus.append(emu)
        scraper.start(interval=0.5)
        try:
            for rate in args.stages:
                import threading

                threads = []
                for v, emu in enumerate(emus):
                    r = rate if v == 0 else 2.0  # extras: steady 2 rps
                    if r > 0:
                        t = threading.Thread(
                            target=drive_load, args=(emu.base_url, r, args.stage_seconds)
                        )
                        t.start()
                        threads.append(t)
                if not threads:
                    time.sleep(max(args.stage_seconds, 10.0))
                for t in threads:
                    t.join()
                t0 = time.perf_counter()
                rec.reconcile()
                cycle_ms = (time.perf_counter() - t0) * 1000.0
                va = cluster.get(v1alpha1.VariantAutoscaling, "vllm-llama", "default")
                desired = va.status.desired_optimized_alloc.num_replicas
                for name in names:
                    simulate_hpa(cluster, registry, name=name)
                entry = {
                    "offered_rps": rate,
                    "measured_arrival_rpm": float(va.status.current_alloc.load.arrival_rate),
                    "desired_replicas": desired,
                    "cycle_ms": cycle_ms,
                    "metrics_ok": v1alpha1.is_condition_true(va, v1alpha1.TYPE_METRICS_AVAILABLE),
                    "optimized_ok": v1alpha1.is_condition_true(va, v1alpha1.TYPE_OPTIMIZATION_READY),
                }
                if args.variants > 1:
                    entry["extra_desired"] = [
                        cluster.get(v1alpha1.VariantAutoscaling, n, "default").status.desired_optimized_alloc.num_replicas
                        for n in names[1:]
                    ]
                trajectory.append(entry)
                print(json.dumps(entry))
        finally:
            scraper.stop()

    result = {"device": device, "stage_seconds": args.stage_seconds, "trajectory": trajectory}
    # staircase sanity: rises with load, returns to baseline on idle
    peaks = [t["desired_replicas"] for t in trajectory]
    result["monotone_rise"] = peaks[2] >= peaks[1] >= peaks[0] >= 1
    result["returns_to_min"] = peaks[-1] == 1
    print(json.dumps({"summary": {k: result[k] for k in ("device", "monotone_rise", "returns_to_min")}}))
    if args.out:
        with open(args.out, "w") as f:
            json.dump(result, f, indent=2)


if __name__ == "__main__":
    main()
