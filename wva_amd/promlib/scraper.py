"""Target scraper: Prometheus text-format -> TimeSeriesStore.

Stands in for the Prometheus server's scrape loop (no reference
counterpart — the reference assumes a cluster Prometheus); adds the
per-target `instance` label exactly as a real scraper would, which is
what keeps multi-replica emulator fleets' series distinct."""

from __future__ import annotations

import threading
import time
from typing import Callable, Dict, List, Optional, Union

from prometheus_client.parser import text_string_to_metric_families

from .store import TimeSeriesStore

Fetcher = Callable[[], Union[str, bytes]]


class Target:
    def __init__(self, fetch: Fetcher, extra_labels: Optional[Dict[str, str]] = None) -> None:
        self.fetch = fetch
        self.extra_labels = extra_labels or {}


class Scraper:
    """Scrapes registered targets into the store.

    Targets are callables returning a /metrics payload; HTTP(S) URLs are
    wrapped into fetchers.  ``extra_labels`` emulates Prometheus relabeling
    (e.g. attaching ``namespace`` the way a ServiceMonitor would).
    """

    def __init__(self, store: TimeSeriesStore) -> None:
        self.store = store
        self._targets: List[Target] = []
        self._thread: Optional[threading.Thread] = None
        self._stop = threading.Event()

    def add_target(self, fetch: Union[Fetcher, str], extra_labels: Optional[Dict[str, str]] = None) -> None:
        labels = dict(extra_labels or {})
        if isinstance(fetch, str):
            url = fetch
            # Prometheus attaches a unique `instance` label per scrape
            # target; without it two pods of one model collide into one
            # series and sum(rate(...)) under-counts the fleet
            labels.setdefault("instance", url.split("//")[-1].split("/")[0])

            def http_fetch() -> bytes:
                import httpx

                return httpx.get(url, timeout=5.0).content

            fetch = http_fetch
        else:
            labels.setdefault("instance", f"target-{len(self._targets)}")
        self._targets.append(Target(fetch, labels))

    def scrape_once(self, ts: Optional[float] = None) -> None:
        ts = time.time() if ts is None else ts
        for target in self._targets:
            try:
                payload = target.fetch()
            except Exception:
                continue
            if isinstance(payload, bytes):
                payload = payload.decode()
            for family in text_string_to_metric_families(payload):
                for sample in family.samples:
                    labels = dict(sample.labels)
                    labels.update(target.extra_labels)
                    self.store.add_sample(sample.name, labels, sample.value, ts)

    def start(self, interval: float = 1.0) -> None:
        self._stop.clear()

        def loop() -> None:
            while not self._stop.is_set():
                self.scrape_once()
                self._stop.wait(interval)

        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5.0)
            self._thread = None
