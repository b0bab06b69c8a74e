"""Serving-loop parameter estimation against the in-repo emulator: the
reference's guidellm two-point procedure
(/root/reference/docs/tutorials/parameter-estimation.md:24-265), run
end-to-end — serve -> measure -> fit -> compare.

Procedure (exactly the tutorial's):

1. **synchronous run** — one request in flight: the measured ITL/TTFT
   are the batch-1 points of the linear laws;
2. **max-concurrency run** — a closed loop holding the server at its
   max batch size: the saturated points;
3. two-point fit:  beta = (ITL_sat - ITL_sync) / (N - 1),
   alpha = ITL_sync - beta, and likewise gamma/delta from TTFT over
   delta * inTokens * batch.

Because the emulator's true step laws are CONFIGURED (alpha, beta,
gamma, delta), the loop validates itself: the fit must recover the
configured parameters within tolerance — the in-repo analog of running
guidellm against a real vLLM server and trusting the resulting profile.

    python tools/profiler/emulator_fit.py --duration 20 --out profiles/emulator_fit.json
"""

from __future__ import annotations

import argparse
import asyncio
import json
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent.parent
sys.path.insert(0, str(ROOT))
sys.path.insert(0, str(ROOT / "tools"))

MODEL = "default/llama-8b"
ALPHA, BETA, GAMMA, DELTA = 12.0, 6.0, 4.0, 0.01
MAX_BATCH = 16
OUT_TOKENS = 25
IN_WORDS = 64


def _hist_means(url: str) -> dict:
    import httpx

    text = httpx.get(f"{url}/metrics", timeout=10.0).text
    out = {}
    for line in text.splitlines():
        for name in ("time_per_output_token_seconds", "time_to_first_token_seconds"):
            if line.startswith(f"vllm:{name}_sum"):
                out[f"{name}_sum"] = float(line.rsplit(" ", 1)[1])
            elif line.startswith(f"vllm:{name}_count"):
                out[f"{name}_count"] = float(line.rsplit(" ", 1)[1])
    return out


def _window_means(before: dict, after: dict) -> dict:
    res = {}
    for name, key in (
        ("itl_ms", "time_per_output_token_seconds"),
        ("ttft_ms", "time_to_first_token_seconds"),
    ):
        ds = after[f"{key}_sum"] - before[f"{key}_sum"]
        dc = after[f"{key}_count"] - before[f"{key}_count"]
        res[name] = ds / dc * 1000.0 if dc > 0 else float("nan")
        res[name.replace("_ms", "_n")] = dc
    return res


async def _closed_loop(url: str, concurrency: int, duration_s: float) -> None:
    import httpx

    prompt = " ".join(["lorem"] * IN_WORDS)
    deadline = time.monotonic() + duration_s
    async with httpx.AsyncClient(timeout=120.0) as client:

        async def worker() -> None:
            while time.monotonic() < deadline:
                try:
                    await client.post(
                        f"{url}/v1/chat/completions",
                        json={
                            "model": MODEL,
                            "messages": [{"role": "user", "content": prompt}],
                        },
                    )
                except httpx.HTTPError:
                    await asyncio.sleep(0.1)

        await asyncio.gather(*(worker() for _ in range(concurrency)))


def run(duration_s: float = 20.0) -> dict:
    from vllm_emulator.engine import EmulatorSettings
    from vllm_emulator.fleet import _Instance

    settings = EmulatorSettings(
        model=MODEL,
        decode_alpha=ALPHA,
        decode_beta=BETA,
        prefill_gamma=GAMMA,
        prefill_delta=DELTA,
        avg_generated_len=OUT_TOKENS,
        tokens_distribution="deterministic",
        max_batch_size=MAX_BATCH,
        realtime=True,
    )
    inst = _Instance(settings)
    url = inst.start()
    try:
        # phase 1: synchronous (concurrency 1)
        before = _hist_means(url)
        asyncio.run(_closed_loop(url, 1, duration_s))
        mid = _hist_means(url)
        sync = _window_means(before, mid)

        # phase 2: max concurrency (the server's max batch size)
        asyncio.run(_closed_loop(url, MAX_BATCH, duration_s))
        after = _hist_means(url)
        sat = _window_means(mid, after)
    finally:
        inst.stop()

    n = float(MAX_BATCH)
    beta = (sat["itl_ms"] - sync["itl_ms"]) / (n - 1.0)
    alpha = sync["itl_ms"] - beta
    # TTFT law: gamma + delta * inTokens * batch
    delta = (sat["ttft_ms"] - sync["ttft_ms"]) / (IN_WORDS * (n - 1.0))
    gamma = sync["ttft_ms"] - delta * IN_WORDS

    def err(est, true):
        return (est - true) / true * 100.0 if true else None

    return {
        "configured": {"alpha": ALPHA, "beta": BETA, "gamma": GAMMA, "delta": DELTA},
        "sync_point": sync,
        "saturated_point": sat,
        "fitted": {"alpha": alpha, "beta": beta, "gamma": gamma, "delta": delta},
        "errors_pct": {
            "alpha": err(alpha, ALPHA),
            "beta": err(beta, BETA),
            "gamma": err(gamma, GAMMA),
            "delta": err(delta, DELTA),
        },
        "duration_s": duration_s,
        "max_batch": MAX_BATCH,
        "in_tokens": IN_WORDS,
    }


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--duration", type=float, default=20.0)
    ap.add_argument("--out", default="")
    args = ap.parse_args()
    result = run(args.duration)
    print(json.dumps(result, indent=2))
    if args.out:
        Path(args.out).parent.mkdir(parents=True, exist_ok=True)
        with open(args.out, "w") as f:
            json.dump(result, f, indent=2)


if __name__ == "__main__":
    main()
