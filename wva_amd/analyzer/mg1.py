"""M/G/1 waiting-time analysis for non-exponential service times.

The serving model everywhere else in this package (and in the reference,
/root/reference/pkg/analyzer) is Markovian: the state-dependent M/M/1/K
chain assumes exponentially distributed service times.  Real LLM request
service times are usually *less* variable than exponential — with
near-deterministic output lengths (structured generation, fixed
max_tokens) the service time approaches a constant, and an M/M/1-based
wait prediction overestimates queueing delay by up to 2x.

This module provides the classical corrections:

- ``pollaczek_khinchine_wait`` — exact M/G/1 mean waiting time
  ``Wq = rho / (1 - rho) * (1 + cs^2) / 2 * E[S]`` (Pollaczek-Khinchine);
  ``cs^2`` is the squared coefficient of variation of service time
  (1 = exponential = M/M/1, 0 = deterministic = M/D/1 with exactly half
  the M/M/1 wait);
- ``service_scv_from_tokens`` — maps output-token-count statistics to a
  service-time SCV under the linear decode law (service time is affine
  in the number of decode passes, so token-count variability IS
  service-time variability; the affine offset shrinks it);
- ``MG1Corrector`` — wraps any M/M/1-family wait prediction with the
  Allen-Cunneen style scaling ``(1 + cs^2) / 2``, which is exact for
  M/G/1 and a standard heuristic for the finite-capacity / state-
  dependent chains used by QueueAnalyzer.

Wiring (round 2): the correction is an OPT-IN analyzer mode.  Set
``WVA_ANALYZER=mg1`` and ``WVA_SERVICE_SCV=<cs^2>`` (default 0.5 in mg1
mode) and every sizing path — ``core.create_allocation`` and the batched
fleet solver — constructs its QueueAnalyzer with that scv, scaling the
predicted waiting times by ``(1 + cs^2)/2`` inside the Size bisections
too.  The default remains the reference's Markovian contract (cs^2 = 1,
identity); sizing under cs^2 = 1 when the workload is near-deterministic
over-provisions — docs/design/mg1-analyzer.md carries measured margins.
"""

from __future__ import annotations

import math
import os
from dataclasses import dataclass

__all__ = [
    "pollaczek_khinchine_wait",
    "service_scv_from_tokens",
    "MG1Corrector",
    "MG1Metrics",
    "configured_scv",
    "auto_scv_enabled",
    "recommended_service_scv",
]


def configured_scv() -> float:
    """The service-time cs^2 selected by environment, read per call so
    tests and operators can flip modes without restarts.

    WVA_ANALYZER: "" / "mm1k" (default, Markovian) or "mg1";
    WVA_SERVICE_SCV: cs^2 override (mg1 mode defaults to 0.5 — halfway
    between deterministic and exponential — when unset).
    """
    mode = os.environ.get("WVA_ANALYZER", "").strip().lower()
    if mode in ("", "mm1k", "mm1"):
        return 1.0
    if mode != "mg1":
        raise ValueError(f"unknown WVA_ANALYZER mode {mode!r} (use 'mm1k' or 'mg1')")
    raw = os.environ.get("WVA_SERVICE_SCV", "").strip()
    if not raw or raw.lower() == "auto":
        # "auto": per-server values are estimated from the serving
        # engine's token histogram (collector.estimate_token_scv) and
        # override this fleet-wide fallback
        return 0.5
    scv = float(raw)
    if scv < 0:
        raise ValueError(f"WVA_SERVICE_SCV must be >= 0, got {scv}")
    return scv


def auto_scv_enabled() -> bool:
    """True when mg1 mode should derive cs^2 per model from the measured
    token histogram (WVA_ANALYZER=mg1 + WVA_SERVICE_SCV=auto)."""
    return (
        os.environ.get("WVA_ANALYZER", "").strip().lower() == "mg1"
        and os.environ.get("WVA_SERVICE_SCV", "").strip().lower() == "auto"
    )


def recommended_service_scv(
    mean_out_tokens: float,
    token_scv: float,
    alpha: float,
    beta: float,
    gamma: float,
    delta: float,
    in_tokens: float,
) -> float:
    """Service-time cs^2 from measured token-count variability under the
    batch-1 service law S = gamma + delta*in + (K-1)(alpha + beta): the
    affine offset damps the token SCV (service_scv_from_tokens).  An
    approximation (batch-dependent terms shift mean and deviation the
    same way to first order), clamped to [0, 1] so auto mode never sizes
    less conservatively than exponential."""
    scv = service_scv_from_tokens(
        mean_out_tokens=max(mean_out_tokens - 1.0, 0.0),
        scv_out_tokens=max(token_scv, 0.0),
        decode_time=alpha + beta,
        fixed_time=gamma + delta * max(in_tokens, 0.0),
    )
    return min(max(scv, 0.0), 1.0)


def pollaczek_khinchine_wait(arrival_rate: float, service_time: float, scv: float) -> float:
    """Mean M/G/1 waiting time (time in queue, excluding service).

    arrival_rate and service_time in consistent units (e.g. req/ms and
    ms); scv is Var[S]/E[S]^2.  Raises ValueError when the queue is
    unstable (rho >= 1) or inputs are out of range.
    """
    if arrival_rate < 0 or service_time <= 0 or scv < 0:
        raise ValueError("arrival_rate >= 0, service_time > 0, scv >= 0 required")
    rho = arrival_rate * service_time
    if rho >= 1.0:
        raise ValueError(f"unstable queue: rho = {rho:.4f} >= 1")
    if rho == 0.0:
        return 0.0
    return rho / (1.0 - rho) * (1.0 + scv) / 2.0 * service_time


def service_scv_from_tokens(
    mean_out_tokens: float,
    scv_out_tokens: float,
    decode_time: float,
    fixed_time: float = 0.0,
) -> float:
    """Service-time SCV under the linear law S = fixed + tokens * decode.

    With S affine in the token count, Var[S] = decode^2 * Var[tokens],
    so the token SCV is damped by the square of the variable fraction:
    scv_S = scv_tokens * (tokens*decode / (fixed + tokens*decode))^2.
    fixed_time covers prefill plus the per-request constant (alpha) part
    of the decode law.
    """
    if mean_out_tokens < 0 or scv_out_tokens < 0 or decode_time < 0 or fixed_time < 0:
        raise ValueError("all inputs must be non-negative")
    variable = mean_out_tokens * decode_time
    total = fixed_time + variable
    if total == 0.0:
        return 0.0
    frac = variable / total
    return scv_out_tokens * frac * frac


@dataclass
class MG1Metrics:
    wait: float        # corrected mean waiting time
    markovian_wait: float  # the cs^2 = 1 prediction it was derived from
    scv: float
    correction: float  # (1 + scv) / 2


class MG1Corrector:
    """Allen-Cunneen style wait scaling for M/M/1-family predictions.

    ``correct(w)`` rescales a Markovian mean-wait prediction ``w`` by
    ``(1 + cs^2) / 2`` — exact for M/G/1 (both are ``rho/(1-rho)*E[S]``
    up to that factor), a standard approximation for the state-dependent
    finite chain.  cs^2 = 1 is the identity.
    """

    def __init__(self, scv: float) -> None:
        if scv < 0:
            raise ValueError("scv must be >= 0")
        self.scv = scv
        self.correction = (1.0 + scv) / 2.0

    def correct(self, markovian_wait: float) -> MG1Metrics:
        if markovian_wait < 0 or not math.isfinite(markovian_wait):
            raise ValueError("markovian_wait must be finite and >= 0")
        return MG1Metrics(
            wait=markovian_wait * self.correction,
            markovian_wait=markovian_wait,
            scv=self.scv,
            correction=self.correction,
        )
