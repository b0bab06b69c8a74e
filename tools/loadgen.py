"""Poisson open-loop load generator for the emulator / vLLM endpoints.

Counterpart of /root/reference/tools/vllm-emulator/loadgen.py: fires
chat-completion requests at exponentially distributed inter-arrival times
following a piecewise-constant rate schedule, without waiting for
responses (open loop).
"""

from __future__ import annotations

import argparse
import asyncio
import random
import time
from dataclasses import dataclass
from typing import List, Optional, Sequence


@dataclass
class Stage:
    rate_rps: float
    duration_s: float


class PoissonLoadGenerator:
    def __init__(
        self,
        base_url: str,
        stages: Sequence[Stage],
        *,
        prompt_words: int = 64,
        model: str = "default/llama-8b",
        seed: Optional[int] = None,
    ) -> None:
        self.base_url = base_url.rstrip("/")
        self.stages = list(stages)
        self.prompt = " ".join(["tok"] * prompt_words)
        self.model = model
        self.rng = random.Random(seed)
        self.sent = 0
        self.completed = 0
        self.errors = 0

    async def _fire(self, client) -> None:
        self.sent += 1
        try:
            resp = await client.post(
                f"{self.base_url}/v1/chat/completions",
                json={
                    "model": self.model,
                    "messages": [{"role": "user", "content": self.prompt}],
                },
                timeout=120.0,
            )
            if resp.status_code == 200:
                self.completed += 1
            else:
                self.errors += 1
        except Exception:
            self.errors += 1

    async def run(self) -> None:
        import httpx

        async with httpx.AsyncClient() as client:
            pending: List[asyncio.Task] = []
            for stage in self.stages:
                end = time.monotonic() + stage.duration_s
                while time.monotonic() < end:
                    if stage.rate_rps <= 0:
                        await asyncio.sleep(min(0.1, end - time.monotonic()))
                        continue
                    pending.append(asyncio.create_task(self._fire(client)))
                    await asyncio.sleep(self.rng.expovariate(stage.rate_rps))
            if pending:
                await asyncio.gather(*pending, return_exceptions=True)


def parse_schedule(spec: str) -> List[Stage]:
    """Parse 'rate:duration,rate:duration' (rps:seconds)."""
    stages = []
    for part in spec.split(","):
        rate, dur = part.split(":")
        stages.append(Stage(rate_rps=float(rate), duration_s=float(dur)))
    return stages


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--url", default="http://127.0.0.1:8000")
    ap.add_argument("--schedule", default="8:60,16:60,24:60,16:60,8:60,0:60",
                    help="rate_rps:duration_s comma-separated stages")
    ap.add_argument("--prompt-words", type=int, default=64)
    ap.add_argument("--model", default="default/llama-8b")
    args = ap.parse_args()
    gen = PoissonLoadGenerator(
        args.url, parse_schedule(args.schedule), prompt_words=args.prompt_words, model=args.model
    )
    asyncio.run(gen.run())
    print(f"sent={gen.sent} completed={gen.completed} errors={gen.errors}")


if __name__ == "__main__":
    main()
