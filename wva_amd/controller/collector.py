"""Prometheus collector: vLLM serving metrics + MI355X GPU telemetry.

Parity with /root/reference/internal/collector/collector.go:
- availability validation with emulator fallback (namespace-less query) and
  a 5-minute staleness gate (collector.go:87-156);
- the five 1-minute-rate PromQL shapes (collector.go:170-209) with the
  req/s->req/min and s->ms unit conversions (collector.go:217,233,239);
- NaN/Inf values fixed to 0; maxBatch hardcoded 256 pending server-reported
  values (collector.go:258-259).

MI355X addition: ``collect_gpu_telemetry`` scrapes amd-smi/rocm-smi
exporter series (utilization, VRAM, power) as auxiliary signals — there is
no NVML/DCGM code path anywhere.
"""

from __future__ import annotations

import math
import time
from dataclasses import dataclass
from typing import Dict, Optional

from ..api import v1alpha1
from ..kube import Deployment
from . import constants
from .logger import log
from .promclient import PromAPI, PromQueryError

STALENESS_LIMIT_SECONDS = 5 * 60

# vendor resource prefixes for (future) limited-mode inventory; AMD first
VENDORS = ["amd.com", "nvidia.com", "intel.com"]


def rate_window() -> str:
    """PromQL rate window for the load queries.  The reference hardcodes
    [1m] (collector.go:170-209); here it is configurable via
    WVA_RATE_WINDOW for fast-cadence deployments and tests."""
    import os

    return os.environ.get("WVA_RATE_WINDOW", "1m")


@dataclass
class MetricsValidationResult:
    available: bool
    reason: str
    message: str


def _fix_value(x: float) -> float:
    return 0.0 if (math.isnan(x) or math.isinf(x)) else x


def _query_value(prom: PromAPI, query: str, metric_name: str) -> float:
    try:
        vec = prom.query(query)
    except PromQueryError as e:
        raise PromQueryError(f"failed to query Prometheus for {metric_name}: {e}") from e
    if not vec:
        return 0.0
    return _fix_value(vec[0].value)


# ------------------------------------------------------------ query builders
def arrival_query(model: str, namespace: str) -> str:
    return (
        f'sum(rate({constants.VLLM_REQUEST_SUCCESS_TOTAL}'
        f'{{{constants.LABEL_MODEL_NAME}="{model}",{constants.LABEL_NAMESPACE}="{namespace}"}}[{rate_window()}]))'
    )


def _ratio_query(sum_metric: str, count_metric: str, model: str, namespace: str) -> str:
    sel = f'{{{constants.LABEL_MODEL_NAME}="{model}",{constants.LABEL_NAMESPACE}="{namespace}"}}'
    w = rate_window()
    return f"sum(rate({sum_metric}{sel}[{w}]))/sum(rate({count_metric}{sel}[{w}]))"


def avg_prompt_tokens_query(model: str, namespace: str) -> str:
    return _ratio_query(
        constants.VLLM_REQUEST_PROMPT_TOKENS_SUM,
        constants.VLLM_REQUEST_PROMPT_TOKENS_COUNT,
        model,
        namespace,
    )


def avg_generation_tokens_query(model: str, namespace: str) -> str:
    return _ratio_query(
        constants.VLLM_REQUEST_GENERATION_TOKENS_SUM,
        constants.VLLM_REQUEST_GENERATION_TOKENS_COUNT,
        model,
        namespace,
    )


def ttft_query(model: str, namespace: str) -> str:
    return _ratio_query(
        constants.VLLM_TIME_TO_FIRST_TOKEN_SECONDS_SUM,
        constants.VLLM_TIME_TO_FIRST_TOKEN_SECONDS_COUNT,
        model,
        namespace,
    )


def itl_query(model: str, namespace: str) -> str:
    return _ratio_query(
        constants.VLLM_TIME_PER_OUTPUT_TOKEN_SECONDS_SUM,
        constants.VLLM_TIME_PER_OUTPUT_TOKEN_SECONDS_COUNT,
        model,
        namespace,
    )


# ------------------------------------------------------------- availability
def validate_metrics_availability(
    prom: PromAPI, model_name: str, namespace: str
) -> MetricsValidationResult:
    """Check that vLLM metrics exist and are fresh for (model, namespace)."""
    test_query = (
        f'{constants.VLLM_REQUEST_SUCCESS_TOTAL}'
        f'{{{constants.LABEL_MODEL_NAME}="{model_name}",{constants.LABEL_NAMESPACE}="{namespace}"}}'
    )
    try:
        vec = prom.query(test_query)
    except PromQueryError as e:
        log.error("Error querying Prometheus for metrics validation", model=model_name, error=str(e))
        return MetricsValidationResult(
            False, v1alpha1.REASON_PROMETHEUS_ERROR, f"Failed to query Prometheus: {e}"
        )

    if not vec:
        # emulator fallback: no namespace label
        fallback = (
            f'{constants.VLLM_REQUEST_SUCCESS_TOTAL}'
            f'{{{constants.LABEL_MODEL_NAME}="{model_name}"}}'
        )
        try:
            vec = prom.query(fallback)
        except PromQueryError as e:
            return MetricsValidationResult(
                False, v1alpha1.REASON_PROMETHEUS_ERROR, f"Failed to query Prometheus: {e}"
            )
        if not vec:
            return MetricsValidationResult(
                False,
                v1alpha1.REASON_METRICS_MISSING,
                f"No vLLM metrics found for model '{model_name}' in namespace "
                f"'{namespace}'. Check: (1) ServiceMonitor exists, (2) selector "
                f"matches the vLLM service, (3) vLLM pods expose /metrics, "
                f"(4) Prometheus scrapes the monitoring namespace",
            )

    now = time.time()
    for sample in vec:
        ts = sample.timestamp or now
        age = now - ts
        if age > STALENESS_LIMIT_SECONDS:
            return MetricsValidationResult(
                False,
                v1alpha1.REASON_METRICS_STALE,
                f"vLLM metrics for model '{model_name}' are stale "
                f"(last update: {age:.0f}s ago)",
            )
    return MetricsValidationResult(
        True, v1alpha1.REASON_METRICS_FOUND, "vLLM metrics are available and up-to-date"
    )


# ----------------------------------------------------------------- main path
ACCELERATOR_LABEL = "inference.optimization/acceleratorName"

# TODO(parity): collect the live max batch size from the server
# (collector.go:258-259 hardcodes 256 with the same TODO)
DEFAULT_MAX_BATCH = 256


def add_metrics_to_opt_status(
    va: v1alpha1.VariantAutoscaling,
    deployment: Deployment,
    accelerator_cost: float,
    prom: PromAPI,
) -> v1alpha1.Allocation:
    """Scrape the five vLLM signals and build status.currentAlloc."""
    namespace = deployment.namespace
    model = va.spec.model_id

    arrival = _query_value(prom, arrival_query(model, namespace), "ArrivalRate") * 60.0
    avg_in = _query_value(prom, avg_prompt_tokens_query(model, namespace), "AvgInputTokens")
    avg_out = _query_value(prom, avg_generation_tokens_query(model, namespace), "AvgOutputTokens")
    ttft_ms = _query_value(prom, ttft_query(model, namespace), "TTFTAverageTime") * 1000.0
    itl_ms = _query_value(prom, itl_query(model, namespace), "ITLAverage") * 1000.0

    num_replicas = int(deployment.spec.replicas or 0)
    acc = va.metadata.labels.get(ACCELERATOR_LABEL, "")
    if not acc:
        log.warn("acceleratorName label not found on VariantAutoscaling object", name=va.name)
    cost = num_replicas * accelerator_cost

    return v1alpha1.Allocation(
        accelerator=acc,
        numReplicas=num_replicas,
        maxBatch=DEFAULT_MAX_BATCH,
        variantCost=f"{cost:.2f}",
        ttftAverage=f"{ttft_ms:.2f}",
        itlAverage=f"{itl_ms:.2f}",
        load=v1alpha1.LoadProfile(
            arrivalRate=f"{arrival:.2f}",
            avgInputTokens=f"{avg_in:.2f}",
            avgOutputTokens=f"{avg_out:.2f}",
        ),
    )


# ----------------------------------------------------- MI355X GPU telemetry
@dataclass
class GpuTelemetry:
    utilization_pct: float = 0.0
    vram_used_bytes: float = 0.0
    power_watts: float = 0.0


def collect_gpu_telemetry(prom: PromAPI, namespace: str) -> Optional[GpuTelemetry]:
    """Auxiliary amd-smi exporter signals (best-effort; None when absent)."""
    sel = f'{{{constants.LABEL_NAMESPACE}="{namespace}"}}'
    try:
        util = _query_value(prom, f"avg({constants.AMD_SMI_GPU_UTILIZATION}{sel})", "GpuUtil")
        vram = _query_value(prom, f"sum({constants.AMD_SMI_GPU_VRAM_USED}{sel})", "GpuVram")
        power = _query_value(prom, f"sum({constants.AMD_SMI_GPU_POWER}{sel})", "GpuPower")
    except PromQueryError:
        return None
    return GpuTelemetry(utilization_pct=util, vram_used_bytes=vram, power_watts=power)


def collect_inventory_k8s(client) -> Dict[str, Dict[str, object]]:
    """Live GPU inventory from the cluster's Nodes — a working
    implementation of what the reference stubs out (collector.go:37-42).

    Per node, for each vendor prefix (AMD first), reads the extended-
    resource count ``<vendor>/gpu`` from status.allocatable (falling
    back to status.capacity) and the product from the
    ``<vendor>/gpu.product`` label — the convention installed by
    deploy/kind-emulator/setup.sh and the AMD GPU operator.  Returns
    ``{product: {"count": units, "nodes": n, "vendor": prefix}}``;
    limited mode turns this into the solver's capacity pool (the
    product name matches the accelerator entry's name in the unit-cost
    ConfigMap, e.g. "MI355X")."""
    from ..kube import Node

    out: Dict[str, Dict[str, object]] = {}
    try:
        nodes = client.list(Node)
    except Exception as e:
        log.warn("node inventory unavailable", error=str(e))
        return {}
    for node in nodes:
        for vendor in VENDORS:
            raw = node.status.allocatable.get(f"{vendor}/gpu") or node.status.capacity.get(
                f"{vendor}/gpu"
            )
            if not raw:
                continue
            try:
                count = int(raw)
            except ValueError:
                log.warn("unparseable gpu capacity on node", node=node.name, value=raw)
                continue
            if count <= 0:
                continue
            product = node.metadata.labels.get(f"{vendor}/gpu.product", "")
            if not product:
                log.warn("node has gpus but no product label; skipping",
                         node=node.name, vendor=vendor)
                continue
            entry = out.setdefault(product, {"count": 0, "nodes": 0, "vendor": vendor})
            entry["count"] += count
            entry["nodes"] += 1
    return out


# --------------------------------------------------- token-variance estimate
def token_scv_query(model: str, namespace: str) -> str:
    """Per-le bucket rates of the generation-token histogram."""
    return (
        f'rate(vllm:request_generation_tokens_bucket'
        f'{{{constants.LABEL_MODEL_NAME}="{model}",'
        f'{constants.LABEL_NAMESPACE}="{namespace}"}}[{rate_window()}])'
    )


def estimate_token_scv(prom: PromAPI, model: str, namespace: str):
    """Squared coefficient of variation of output-token counts, estimated
    from the ``vllm:request_generation_tokens`` histogram the serving
    engine already exports (the reference's emulator omits this series;
    ours emits it).

    Cumulative bucket rates -> per-bucket probabilities -> first/second
    moments with bucket midpoints (+Inf capped at 1.5x the last finite
    edge).  Returns None when the histogram is absent or the window is
    empty — callers fall back to the configured cs^2.  This is what
    makes ``WVA_ANALYZER=mg1`` + ``WVA_SERVICE_SCV=auto`` self-
    configuring: measured length variability, not an operator guess.
    """
    try:
        vec = prom.query(token_scv_query(model, namespace))
    except PromQueryError:
        return None
    if not vec:
        # emulator fallback: no namespace label
        try:
            vec = prom.query(
                f'rate(vllm:request_generation_tokens_bucket'
                f'{{{constants.LABEL_MODEL_NAME}="{model}"}}[{rate_window()}])'
            )
        except PromQueryError:
            return None
    buckets = []
    for s in vec:
        le = s.labels.get("le", "")
        if not le:
            continue
        edge = math.inf if le in ("+Inf", "Inf", "inf") else float(le)
        buckets.append((edge, max(_fix_value(s.value), 0.0)))
    if len(buckets) < 2:
        return None
    buckets.sort(key=lambda b: b[0])
    total = buckets[-1][1]
    if total <= 0:
        return None
    last_finite = max((e for e, _ in buckets if math.isfinite(e)), default=0.0)
    m1 = 0.0
    m2 = 0.0
    prev_edge, prev_cum = 0.0, 0.0
    for edge, cum in buckets:
        p = max(cum - prev_cum, 0.0) / total
        hi = edge if math.isfinite(edge) else last_finite * 1.5
        mid = 0.5 * (prev_edge + hi)
        m1 += p * mid
        m2 += p * mid * mid
        prev_edge, prev_cum = (hi if math.isfinite(edge) else prev_edge), cum
    if m1 <= 0:
        return None
    var = max(m2 - m1 * m1, 0.0)
    return var / (m1 * m1)
