"""Quantify the M/M/1 over-provisioning margin on low-variability
workloads: drive the emulator with DETERMINISTIC output
lengths (cs^2 ~ 0) and with the heavy-tailed ShareGPT-like distribution
(cs^2 ~ 0.9), measure the actual queueing delay from the
``vllm:request_queue_time_seconds`` histogram, and compare it against
the analyzer's wait prediction under cs^2 = 1 (the Markovian default)
and under the Allen-Cunneen corrected value.

The expectation being validated: for deterministic lengths the measured
wait sits near HALF the M/M/1 prediction (the M/D/1 limit), i.e. sizing
that workload with cs^2 = 1 over-provisions; for the heavy-tailed trace
the Markovian prediction is about right.  Results feed
docs/design/mg1-analyzer.md.

    python tools/mg1_experiment.py --rate 4.0 --duration 30 --out profiles/r02_mg1_experiment.json
"""

from __future__ import annotations

import argparse
import asyncio
import json
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))
sys.path.insert(0, str(ROOT / "tools"))

MODEL = "default/llama-8b"
ALPHA, BETA, GAMMA, DELTA = 12.0, 6.0, 4.0, 0.01
MAX_BATCH = 16
OUT_TOKENS = 25
IN_WORDS = 32


def run_scenario(distribution: str, rate_rps: float, duration_s: float) -> dict:
    import httpx

    from vllm_emulator.engine import EmulatorSettings
    from vllm_emulator.fleet import _Instance
    from loadgen import PoissonLoadGenerator, Stage

    settings = EmulatorSettings(
        model=MODEL,
        decode_alpha=ALPHA,
        decode_beta=BETA,
        prefill_gamma=GAMMA,
        prefill_delta=DELTA,
        avg_generated_len=OUT_TOKENS,
        tokens_distribution=distribution,
        max_batch_size=MAX_BATCH,
        realtime=True,
    )
    inst = _Instance(settings)
    url = inst.start()
    try:
        gen = PoissonLoadGenerator(
            url,
            [Stage(rate_rps, duration_s)],
            prompt_words=IN_WORDS,
            model=MODEL,
            seed=7,
        )
        asyncio.run(gen.run())
        time.sleep(0.5)
        text = httpx.get(f"{url}/metrics", timeout=10.0).text
    finally:
        inst.stop()

    # parse histogram sum/count lines
    def series(name: str, suffix: str) -> float:
        for line in text.splitlines():
            if line.startswith(f'vllm:{name}_{suffix}'):
                return float(line.rsplit(" ", 1)[1])
        return float("nan")

    gen_sum = series("request_generation_tokens", "sum")
    gen_cnt = max(series("request_generation_tokens", "count"), 1.0)
    mean_tokens = gen_sum / gen_cnt
    # mean wait over ALL completed requests (non-waiters contribute 0),
    # matching the analyzer's definition of mean waiting time
    wait_ms = series("request_queue_time_seconds", "sum") / gen_cnt * 1000.0
    return {
        "distribution": distribution,
        "offered_rps": rate_rps,
        "completed": gen_cnt,
        "waiters": series("request_queue_time_seconds", "count"),
        "mean_out_tokens": mean_tokens,
        "measured_wait_ms": wait_ms,
    }


def predictions(rate_rps: float) -> dict:
    from wva_amd.analyzer import (
        Configuration,
        DecodeParms,
        PrefillParms,
        QueueAnalyzer,
        RequestSize,
        ServiceParms,
    )

    config = Configuration(
        max_batch_size=MAX_BATCH,
        max_queue_size=MAX_BATCH * 10,
        service_parms=ServiceParms(
            prefill=PrefillParms(gamma=GAMMA, delta=DELTA),
            decode=DecodeParms(alpha=ALPHA, beta=BETA),
        ),
    )
    rs = RequestSize(avg_input_tokens=IN_WORDS, avg_output_tokens=OUT_TOKENS)
    out = {}
    for scv in (1.0, 0.5, 0.0):
        qa = QueueAnalyzer(config, rs, scv=scv)
        out[f"predicted_wait_ms_scv{scv:g}"] = qa.analyze(rate_rps).avg_wait_time
    return out


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--rate", type=float, default=5.2)
    ap.add_argument("--duration", type=float, default=60.0)
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    pred = predictions(args.rate)
    rows = []
    for dist in ("deterministic", "sharegpt"):
        r = run_scenario(dist, args.rate, args.duration)
        r.update(pred)
        mm1 = pred["predicted_wait_ms_scv1"]
        md1 = pred["predicted_wait_ms_scv0"]
        obs = r["measured_wait_ms"]
        r["err_vs_mm1_pct"] = (mm1 - obs) / obs * 100.0 if obs else None
        r["err_vs_md1_pct"] = (md1 - obs) / obs * 100.0 if obs else None
        rows.append(r)
        print(json.dumps(r))

    result = {"rate_rps": args.rate, "duration_s": args.duration, "scenarios": rows}
    if args.out:
        Path(args.out).parent.mkdir(parents=True, exist_ok=True)
        with open(args.out, "w") as f:
            json.dump(result, f, indent=2)


if __name__ == "__main__":
    main()
