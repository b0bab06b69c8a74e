"""Inference-server queue analyzer: build, analyze, size.

Parity with /root/reference/pkg/analyzer/queueanalyzer.go:

- service time of a batch of n requests:
    prefill(n) = gamma + delta * inTokens * n       (0 when inTokens == 0)
    decode(n)  = alpha + beta * n
    state-dependent service rate mu(n) = n / (prefill(n) + numDecode*decode(n))
  with numDecode = outTokens-1, special-cased to 1 for decode-only/one-token
  requests (queueanalyzer.go:104-110);
- stable arrival-rate range [eps*mu(1), (1-eps)*mu(N)], eps = 0.001;
- occupancy bound K = maxQueueSize + maxBatchSize;
- Analyze(rate) evaluates throughput/latency/utilization at a rate (req/s);
- Size(targets) inverts the model: binary search for the max rate meeting
  TTFT and ITL targets, TPS handled as lambdaMax*(1 - 0.1) without a search
  (queueanalyzer.go:231-234), final rate = min of the three.

Units: rates are req/s at the public boundary, req/ms internally; latencies
are milliseconds throughout.  No global evaluation state — eval functions are
closures over this instance.
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np

from .search import BelowRegion, binary_search
from .statedep import MM1ModelStateDependent

# small disturbance around a value
EPSILON = 0.001
# run this fraction below maximum server throughput for stability
STABILITY_SAFETY_FRACTION = 0.1


class AnalyzerError(RuntimeError):
    """Invalid configuration, request size, target, or model evaluation."""


@dataclass(frozen=True)
class PrefillParms:
    """prefill time = gamma + delta * inputTokens * batchSize (ms)."""

    gamma: float
    delta: float

    def prefill_time(self, avg_input_tokens: int, batch_size: float) -> float:
        if avg_input_tokens == 0:
            return 0.0
        return self.gamma + self.delta * avg_input_tokens * batch_size


@dataclass(frozen=True)
class DecodeParms:
    """decode time = alpha + beta * batchSize (ms)."""

    alpha: float
    beta: float

    def decode_time(self, batch_size: float) -> float:
        return self.alpha + self.beta * batch_size


@dataclass(frozen=True)
class ServiceParms:
    prefill: PrefillParms
    decode: DecodeParms


@dataclass(frozen=True)
class RequestSize:
    avg_input_tokens: int
    avg_output_tokens: int

    def check(self) -> None:
        if self.avg_input_tokens < 0 or self.avg_output_tokens < 1:
            raise AnalyzerError(f"invalid request size {self}")


@dataclass(frozen=True)
class Configuration:
    max_batch_size: int
    max_queue_size: int
    service_parms: ServiceParms

    def check(self) -> None:
        if self.max_batch_size <= 0 or self.max_queue_size < 0 or self.service_parms is None:
            raise AnalyzerError(f"invalid configuration {self}")


@dataclass(frozen=True)
class RateRange:
    """Stable request-rate range in req/s."""

    min: float
    max: float


@dataclass(frozen=True)
class AnalysisMetrics:
    throughput: float  # effective throughput (req/s)
    avg_resp_time: float  # average request latency (ms)
    avg_wait_time: float  # average queueing time (ms)
    avg_num_in_serv: float  # average number of requests in service
    avg_prefill_time: float  # average prefill time (ms)
    avg_token_time: float  # average token decode time (ms)
    max_rate: float  # maximum throughput (req/s)
    rho: float  # utilization


@dataclass(frozen=True)
class TargetPerf:
    target_ttft: float = 0.0  # queueing + prefill (ms); 0 disables
    target_itl: float = 0.0  # inter-token latency (ms); 0 disables
    target_tps: float = 0.0  # token throughput (tok/s); 0 disables

    def check(self) -> None:
        if self.target_itl < 0 or self.target_ttft < 0 or self.target_tps < 0:
            raise AnalyzerError(f"invalid target data values {self}")


@dataclass(frozen=True)
class TargetRate:
    rate_target_ttft: float  # req/s
    rate_target_itl: float  # req/s
    rate_target_tps: float  # req/s


def effective_concurrency(
    avg_service_time: float,
    service_parms: ServiceParms,
    request_size: RequestSize,
    max_batch_size: int,
) -> float:
    """Effective average number of requests in service: solve
    prefill(n) + (outTokens-1)*decode(n) = avgServiceTime for n,
    clamped to [0, maxBatchSize] (queueanalyzer.go:296-302).
    """
    tokens = float(request_size.avg_output_tokens - 1)
    numerator = avg_service_time - (service_parms.prefill.gamma + service_parms.decode.alpha * tokens)
    denominator = service_parms.prefill.delta * request_size.avg_input_tokens + service_parms.decode.beta * tokens
    if denominator == 0.0:
        # degenerate linear model (e.g. one output token with no prefill
        # slope): any n satisfies the identity up to the constant term
        return float(max_batch_size) if numerator > 0 else 0.0
    n = numerator / denominator
    return min(max(n, 0.0), float(max_batch_size))


def build_service_rates(config: Configuration, request_size: RequestSize) -> np.ndarray:
    """State-dependent service rates mu(n), n = 1..maxBatchSize, in req/ms."""
    parms = config.service_parms
    n = np.arange(1, config.max_batch_size + 1, dtype=np.float64)
    if request_size.avg_input_tokens == 0:
        prefill = np.zeros_like(n)
    else:
        prefill = parms.prefill.gamma + parms.prefill.delta * request_size.avg_input_tokens * n
    num_decode = request_size.avg_output_tokens - 1
    if request_size.avg_input_tokens == 0 and request_size.avg_output_tokens == 1:
        num_decode = 1
    decode = num_decode * (parms.decode.alpha + parms.decode.beta * n)
    return n / (prefill + decode)


class QueueAnalyzer:
    """Analyzer of one inference-server queue for a fixed (model, accelerator,
    request-shape) point.

    ``scv`` (squared coefficient of variation of service time) selects the
    queueing model family: 1.0 (default) is the reference's Markovian
    M/M/1/K contract; any other value applies the Allen-Cunneen wait
    scaling ``(1 + cs^2)/2`` (exact for M/G/1) to every waiting-time
    prediction — inside the Size bisections too, so lambda* and replica
    counts respond.  Opt in fleet-wide with ``WVA_ANALYZER=mg1`` +
    ``WVA_SERVICE_SCV`` (see analyzer/mg1.py for the theory and
    docs/design/mg1-analyzer.md for measured margins)."""

    def __init__(
        self, config: Configuration, request_size: RequestSize, scv: float = 1.0
    ) -> None:
        config.check()
        request_size.check()
        if scv < 0:
            raise AnalyzerError(f"invalid service-time scv {scv}")
        self.max_batch_size = config.max_batch_size
        self.max_queue_size = config.max_queue_size
        self.service_parms = config.service_parms
        self.request_size = request_size
        self.scv = scv
        self._wait_scale = (1.0 + scv) / 2.0

        serv_rate = build_service_rates(config, request_size)
        self.serv_rate = serv_rate
        lambda_min = float(serv_rate[0]) * EPSILON
        lambda_max = float(serv_rate[-1]) * (1.0 - EPSILON)
        self.rate_range = RateRange(min=lambda_min * 1000.0, max=lambda_max * 1000.0)
        occupancy_upper_bound = config.max_queue_size + config.max_batch_size
        self.model = MM1ModelStateDependent(occupancy_upper_bound, serv_rate)

    # -- evaluation ---------------------------------------------------------
    def _solve(self, lam_per_ms: float) -> MM1ModelStateDependent:
        self.model.solve(lam_per_ms, 1.0)
        if not self.model.is_valid:
            raise AnalyzerError(f"invalid model {self.model!r}")
        return self.model

    def _eval_ttft(self, lam_per_ms: float) -> float:
        m = self._solve(lam_per_ms)
        eff_conc = effective_concurrency(
            m.avg_serv_time, self.service_parms, self.request_size, self.max_batch_size
        )
        return m.avg_wait_time * self._wait_scale + self.service_parms.prefill.prefill_time(
            self.request_size.avg_input_tokens, eff_conc
        )

    def _eval_itl(self, lam_per_ms: float) -> float:
        m = self._solve(lam_per_ms)
        eff_conc = effective_concurrency(
            m.avg_serv_time, self.service_parms, self.request_size, self.max_batch_size
        )
        return self.service_parms.decode.decode_time(eff_conc)

    # -- public API ---------------------------------------------------------
    def analyze(self, request_rate: float) -> AnalysisMetrics:
        """Evaluate performance metrics at a given request rate (req/s)."""
        if request_rate <= 0:
            raise AnalyzerError(f"invalid request rate {request_rate}")
        if request_rate > self.rate_range.max:
            raise AnalyzerError(
                f"rate={request_rate}, max allowed rate={self.rate_range.max}"
            )
        m = self._solve(request_rate / 1000.0)
        avg_num_in_serv = m.avg_num_in_servers
        eff_conc = effective_concurrency(
            m.avg_serv_time, self.service_parms, self.request_size, self.max_batch_size
        )
        prefill_time = self.service_parms.prefill.prefill_time(
            self.request_size.avg_input_tokens, eff_conc
        )
        token_time = self.service_parms.decode.decode_time(eff_conc)
        rho = min(max(avg_num_in_serv / float(self.max_batch_size), 0.0), 1.0)
        corrected_wait = m.avg_wait_time * self._wait_scale
        # scv == 1 keeps the reference's exact Markovian resp time
        # (including its wait>=0 clamp edge case)
        resp = (
            m.avg_resp_time
            if self._wait_scale == 1.0
            else m.avg_serv_time + corrected_wait
        )
        return AnalysisMetrics(
            throughput=m.throughput * 1000.0,
            avg_resp_time=resp,
            avg_wait_time=corrected_wait,
            avg_num_in_serv=avg_num_in_serv,
            avg_prefill_time=prefill_time,
            avg_token_time=token_time,
            max_rate=self.rate_range.max,
            rho=rho,
        )

    def size(self, target_perf: TargetPerf):
        """Max request rates achieving the targets.

        Returns ``(target_rate, metrics, achieved)`` where ``metrics`` is the
        analysis at the min of the three max rates.  Raises
        :class:`AnalyzerError` when a target lies below the reachable region
        (=> no feasible allocation upstream).
        """
        target_perf.check()
        lambda_min = self.rate_range.min / 1000.0
        lambda_max = self.rate_range.max / 1000.0

        lambda_star_ttft = lambda_max
        if target_perf.target_ttft > 0:
            lambda_star_ttft, ind = binary_search(
                lambda_min, lambda_max, target_perf.target_ttft, self._eval_ttft
            )
            if ind == BelowRegion:
                raise AnalyzerError(
                    f"failed to calculate lambdaStarTTFT: target {target_perf.target_ttft} "
                    f"is below the bounded region {self.rate_range}"
                )

        lambda_star_itl = lambda_max
        if target_perf.target_itl > 0:
            lambda_star_itl, ind = binary_search(
                lambda_min, lambda_max, target_perf.target_itl, self._eval_itl
            )
            if ind == BelowRegion:
                raise AnalyzerError(
                    f"failed to calculate lambdaStarITL: target {target_perf.target_itl} "
                    f"is below the bounded region {self.rate_range}"
                )

        lambda_star_tps = lambda_max
        if target_perf.target_tps > 0:
            lambda_star_tps = lambda_max * (1.0 - STABILITY_SAFETY_FRACTION)

        lam = min(lambda_star_ttft, lambda_star_itl, lambda_star_tps)
        metrics = self.analyze(lam * 1000.0)

        target_rate = TargetRate(
            rate_target_ttft=lambda_star_ttft * 1000.0,
            rate_target_itl=lambda_star_itl * 1000.0,
            rate_target_tps=lambda_star_tps * 1000.0,
        )
        achieved = TargetPerf(
            target_ttft=metrics.avg_wait_time + metrics.avg_prefill_time,
            target_itl=metrics.avg_token_time,
            target_tps=metrics.throughput * self.request_size.avg_output_tokens,
        )
        return target_rate, metrics, achieved
