"""PromAPI adapter over the promlib store — what the collector talks to.

No reference counterpart: the reference's e2e tier needs a live
kube-prometheus-stack (test/utils deploy-llm-d.sh installs one with
self-signed TLS); promlib replaces that dependency in-process so the
same collector queries run offline."""

from __future__ import annotations

import time
from typing import List

from ..controller.promclient import PromQueryError, Sample
from .promql import PromQLError, evaluate
from .store import TimeSeriesStore


class PromlibAPI:
    def __init__(self, store: TimeSeriesStore, now_fn=None) -> None:
        self.store = store
        self.now_fn = now_fn or time.time

    def query(self, query: str) -> List[Sample]:
        if query == "up":
            return [Sample(value=1.0, timestamp=self.now_fn())]
        try:
            return evaluate(query, self.store, now=self.now_fn())
        except PromQLError as e:
            raise PromQueryError(str(e)) from e
