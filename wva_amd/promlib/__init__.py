"""Offline mini-Prometheus: scraper + PromQL-subset engine.

The reference's e2e tier needs a real Prometheus scraping the emulator
(SURVEY.md §4.3).  This environment has no Prometheus binary and no
network egress, so promlib provides the minimum honest equivalent:

- :class:`TimeSeriesStore` — in-memory TSDB of scraped samples;
- :class:`Scraper` — pulls Prometheus text-format /metrics payloads from
  registered targets (callables or HTTP URLs) on a cadence;
- :func:`evaluate` — evaluates the PromQL subset the controller emits:
  instant vector selectors, ``rate(m{...}[w])``, ``sum/avg/min/max(...)``
  aggregations and ratios of two aggregations;
- :class:`PromlibAPI` — the PromAPI-protocol adapter the collector plugs
  into, indistinguishable from a real Prometheus to the controller.
"""

from .store import Series, TimeSeriesStore
from .scraper import Scraper
from .promql import PromQLError, evaluate
from .api import PromlibAPI

__all__ = [
    "Series",
    "TimeSeriesStore",
    "Scraper",
    "PromQLError",
    "evaluate",
    "PromlibAPI",
]
