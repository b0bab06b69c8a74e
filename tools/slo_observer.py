"""Observed-vs-predicted SLO scoring.

The round-1 bench judged SLO attainment with the same analytic model
that sized the allocations — near-tautological.  This module scores a
closed loop the way the reference's hardware e2e does
(/root/reference/test/e2e-openshift/sharegpt_scaleup_test.go): *observed*
serving latency (the emulator fleet's Prometheus TTFT/ITL histograms,
read back through the collector's own query shapes) against the service
-class targets, alongside the analyzer's *prediction* for the chosen
allocation so the model-vs-reality drift is a measured figure.
"""

from __future__ import annotations

import sys
from dataclasses import dataclass
from pathlib import Path
from typing import Optional

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from wva_amd.analyzer import (
    Configuration,
    DecodeParms,
    PrefillParms,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
)
from wva_amd.config import MAX_QUEUE_TO_BATCH_RATIO
from wva_amd.controller.collector import itl_query, ttft_query


@dataclass
class LatencyObservation:
    ttft_ms: float
    itl_ms: float


@dataclass
class SLOScore:
    observed: LatencyObservation
    predicted: Optional[LatencyObservation]
    target_ttft_ms: float
    target_itl_ms: float
    observed_met: bool
    itl_drift_pct: Optional[float]
    ttft_drift_pct: Optional[float]

    def as_dict(self) -> dict:
        return {
            "observed_ttft_ms": self.observed.ttft_ms,
            "observed_itl_ms": self.observed.itl_ms,
            "predicted_ttft_ms": self.predicted.ttft_ms if self.predicted else None,
            "predicted_itl_ms": self.predicted.itl_ms if self.predicted else None,
            "target_ttft_ms": self.target_ttft_ms,
            "target_itl_ms": self.target_itl_ms,
            "observed_met": self.observed_met,
            "itl_drift_pct": self.itl_drift_pct,
            "ttft_drift_pct": self.ttft_drift_pct,
        }


def observe_latency(prom, model: str, namespace: str) -> LatencyObservation:
    """Measured mean TTFT/ITL over the collector's rate window, read with
    the collector's own PromQL shapes (s -> ms at the boundary,
    collector.go:233,239 parity)."""

    def val(query: str) -> float:
        vec = prom.query(query)
        if not vec:
            return float("nan")
        return float(vec[0].value)

    return LatencyObservation(
        ttft_ms=val(ttft_query(model, namespace)) * 1000.0,
        itl_ms=val(itl_query(model, namespace)) * 1000.0,
    )


def predict_latency(
    alpha: float,
    beta: float,
    gamma: float,
    delta: float,
    max_batch: int,
    in_tokens: int,
    out_tokens: int,
    per_replica_rate_rps: float,
) -> Optional[LatencyObservation]:
    """Analyzer prediction for one replica at its share of the load —
    what the sizing model expects the fleet to deliver after actuation."""
    config = Configuration(
        max_batch_size=max_batch,
        max_queue_size=max_batch * MAX_QUEUE_TO_BATCH_RATIO,
        service_parms=ServiceParms(
            prefill=PrefillParms(gamma=gamma, delta=delta),
            decode=DecodeParms(alpha=alpha, beta=beta),
        ),
    )
    try:
        qa = QueueAnalyzer(
            config,
            RequestSize(avg_input_tokens=in_tokens, avg_output_tokens=out_tokens),
        )
        m = qa.analyze(per_replica_rate_rps)
    except Exception:
        return None
    return LatencyObservation(
        ttft_ms=m.avg_wait_time + m.avg_prefill_time, itl_ms=m.avg_token_time
    )


def score(
    observed: LatencyObservation,
    predicted: Optional[LatencyObservation],
    target_ttft_ms: float,
    target_itl_ms: float,
) -> SLOScore:
    met = (
        observed.itl_ms == observed.itl_ms  # not NaN
        and observed.ttft_ms == observed.ttft_ms
        and (target_itl_ms == 0 or observed.itl_ms <= target_itl_ms)
        and (target_ttft_ms == 0 or observed.ttft_ms <= target_ttft_ms)
    )

    def drift(pred: float, obs: float) -> Optional[float]:
        if obs != obs or obs == 0 or pred != pred:
            return None
        return abs(pred - obs) / obs * 100.0

    return SLOScore(
        observed=observed,
        predicted=predicted,
        target_ttft_ms=target_ttft_ms,
        target_itl_ms=target_itl_ms,
        observed_met=met,
        itl_drift_pct=drift(predicted.itl_ms, observed.itl_ms) if predicted else None,
        ttft_drift_pct=drift(predicted.ttft_ms, observed.ttft_ms) if predicted else None,
    )
