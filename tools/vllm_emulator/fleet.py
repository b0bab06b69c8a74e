"""Scalable emulator fleet: N serving instances behind a load-splitting
front, resized like a Deployment.

Closes the actuation loop the reference proves on hardware
(/root/reference/test/e2e-openshift/sharegpt_scaleup_test.go:39-253: WVA
recommends -> HPA scales -> observed serving latency returns under the
SLO).  One emulator process has fixed capacity, so observed TTFT/ITL can
only respond to a scaling decision if the fleet actually grows —
``EmulatorFleet`` pre-starts ``max_replicas`` instances and routes load
round-robin across the first ``replicas`` of them, standing in for the
Deployment + Service the external HPA resizes.
"""

from __future__ import annotations

import asyncio
import threading
import time
from typing import List

from .engine import EmulatorSettings
from .server import create_app


class _Instance:
    """Uvicorn-hosted emulator on an ephemeral localhost port."""

    def __init__(self, settings: EmulatorSettings) -> None:
        import uvicorn

        self.app = create_app(settings)
        self._server = uvicorn.Server(
            uvicorn.Config(self.app, host="127.0.0.1", port=0, log_level="error")
        )
        self._thread = threading.Thread(target=self._server.run, daemon=True)

    def start(self) -> str:
        self._thread.start()
        for _ in range(200):
            if self._server.started:
                break
            time.sleep(0.05)
        assert self._server.started
        port = self._server.servers[0].sockets[0].getsockname()[1]
        self.base_url = f"http://127.0.0.1:{port}"
        return self.base_url

    def stop(self) -> None:
        self._server.should_exit = True
        self._thread.join(timeout=10.0)


class EmulatorFleet:
    """``max_replicas`` emulator instances; load is split across the
    first ``replicas`` (the current Deployment size)."""

    def __init__(self, settings: EmulatorSettings, max_replicas: int = 8) -> None:
        self.settings = settings
        self.instances: List[_Instance] = []
        self.urls: List[str] = []
        self.replicas = 1
        for _ in range(max_replicas):
            inst = _Instance(settings)
            self.instances.append(inst)
            self.urls.append(inst.start())

    def scale(self, replicas: int) -> None:
        self.replicas = max(1, min(replicas, len(self.instances)))

    @property
    def active_urls(self) -> List[str]:
        return self.urls[: self.replicas]

    def drive(self, rate_rps: float, duration_s: float, model: str,
              prompt_words: int = 32, seed: int = 1) -> None:
        """Offered load split evenly over the active instances (the
        Service round-robin analog); blocks until the stage ends."""
        import sys
        from pathlib import Path

        tools_dir = str(Path(__file__).resolve().parent.parent)
        if tools_dir not in sys.path:
            sys.path.insert(0, tools_dir)
        from loadgen import PoissonLoadGenerator, Stage

        active = self.active_urls
        per = rate_rps / len(active)

        def run_one(url: str, worker_seed: int) -> None:
            gen = PoissonLoadGenerator(
                url,
                [Stage(per, duration_s)],
                prompt_words=prompt_words,
                model=model,
                seed=worker_seed,
            )
            asyncio.run(gen.run())

        threads = [
            threading.Thread(target=run_one, args=(url, seed + i))
            for i, url in enumerate(active)
        ]
        for t in threads:
            t.start()
        for t in threads:
            t.join()

    def close(self) -> None:
        for inst in self.instances:
            inst.stop()

    def __enter__(self) -> "EmulatorFleet":
        return self

    def __exit__(self, *exc) -> None:
        self.close()
