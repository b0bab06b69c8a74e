"""Offline batch simulation: run the emulator in virtual time under a
Poisson arrival schedule and report TTFT / latency / queue / memory stats.

Counterpart of the reference's tools/vllm-emulator/experiment.py (which
plots with matplotlib); here results are JSON on stdout so they feed CI
and docs.

    python -m vllm_emulator.experiment --rate 5 --duration 60 \
        --in-tokens 128 --avg-tokens 100
"""

from __future__ import annotations

import argparse
import asyncio
import json
import random
from typing import List

import numpy as np

from .engine import EmulatedVLLM, EmulatorSettings, RequestElement
from .metrics import EmulatorMetrics


async def simulate(
    settings: EmulatorSettings,
    rate_rps: float,
    duration_s: float,
    in_tokens: int,
    seed: int = 0,
) -> dict:
    settings.realtime = False
    metrics = EmulatorMetrics(settings.model)
    engine = EmulatedVLLM(settings, metrics)
    rng = random.Random(seed)
    out_rng = random.Random(seed + 1)

    # pre-draw the Poisson arrival schedule in virtual ms
    arrivals: List[float] = []
    t = 0.0
    while t < duration_s * 1000.0:
        t += rng.expovariate(rate_rps) * 1000.0
        arrivals.append(t)

    requests: List[RequestElement] = []
    i = 0
    max_virtual_ms = duration_s * 1000.0 * 3  # drain margin
    while engine.clock.now_ms < max_virtual_ms:
        while i < len(arrivals) and arrivals[i] <= engine.clock.now_ms:
            gen = max(1, int(out_rng.gauss(settings.avg_generated_len, settings.avg_generated_len / 4)))
            req = RequestElement(f"r{i}", in_tokens, in_tokens + gen)
            engine.submit(req)
            requests.append(req)
            i += 1
        await engine.one_iteration()
        if i >= len(arrivals) and not engine.running and not engine.waiting:
            break

    done = [r for r in requests if r.stage == "finished"]
    ttft = np.array([r.first_token_ms - r.arrival_ms for r in done if r.first_token_ms is not None])
    latency = np.array([r.completion_ms - r.arrival_ms for r in done])
    return {
        "offered_rate_rps": rate_rps,
        "submitted": len(requests),
        "completed": len(done),
        "virtual_seconds": engine.clock.now_ms / 1000.0,
        "throughput_rps": len(done) / (engine.clock.now_ms / 1000.0) if engine.clock.now_ms else 0.0,
        "ttft_ms": {
            "mean": float(ttft.mean()) if ttft.size else None,
            "p50": float(np.percentile(ttft, 50)) if ttft.size else None,
            "p95": float(np.percentile(ttft, 95)) if ttft.size else None,
        },
        "latency_ms": {
            "mean": float(latency.mean()) if latency.size else None,
            "p95": float(np.percentile(latency, 95)) if latency.size else None,
        },
        "peak_kv_usage_frac": metrics.registry.get_sample_value(
            "vllm:gpu_cache_usage_perc", {"model_name": settings.model}
        ),
    }


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--rate", type=float, default=5.0)
    ap.add_argument("--duration", type=float, default=30.0)
    ap.add_argument("--in-tokens", type=int, default=128)
    ap.add_argument("--avg-tokens", type=int, default=100)
    ap.add_argument("--max-batch", type=int, default=256)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    settings = EmulatorSettings(
        avg_generated_len=args.avg_tokens, max_batch_size=args.max_batch, realtime=False
    )
    result = asyncio.run(
        simulate(settings, args.rate, args.duration, args.in_tokens, args.seed)
    )
    print(json.dumps(result, indent=2))


if __name__ == "__main__":
    main()
