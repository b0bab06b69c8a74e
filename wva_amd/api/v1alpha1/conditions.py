"""Status condition types, reasons and helpers.

Parity with /root/reference/api/v1alpha1/conditions.go and the
k8s.io/apimachinery ``meta.SetStatusCondition`` semantics it relies on:
LastTransitionTime only changes when the condition *status* changes.
"""

from __future__ import annotations

import datetime
from typing import Optional

from .types import Condition, VariantAutoscaling

# condition types
TYPE_METRICS_AVAILABLE = "MetricsAvailable"
TYPE_OPTIMIZATION_READY = "OptimizationReady"

# MetricsAvailable reasons
REASON_METRICS_FOUND = "MetricsFound"
REASON_METRICS_MISSING = "MetricsMissing"
REASON_METRICS_STALE = "MetricsStale"
REASON_PROMETHEUS_ERROR = "PrometheusError"

# OptimizationReady reasons
REASON_OPTIMIZATION_SUCCEEDED = "OptimizationSucceeded"
REASON_OPTIMIZATION_FAILED = "OptimizationFailed"
REASON_METRICS_UNAVAILABLE = "MetricsUnavailable"


def _now() -> datetime.datetime:
    return datetime.datetime.now(datetime.timezone.utc)


def set_condition(
    va: VariantAutoscaling, condition_type: str, status: str, reason: str, message: str
) -> None:
    """Set or update a condition (meta.SetStatusCondition semantics)."""
    new = Condition(
        type=condition_type,
        status=status,
        observedGeneration=va.metadata.generation,
        lastTransitionTime=_now(),
        reason=reason,
        message=message,
    )
    for i, existing in enumerate(va.status.conditions):
        if existing.type == condition_type:
            if existing.status == status:
                new.last_transition_time = existing.last_transition_time
            va.status.conditions[i] = new
            return
    va.status.conditions.append(new)


def get_condition(va: VariantAutoscaling, condition_type: str) -> Optional[Condition]:
    for c in va.status.conditions:
        if c.type == condition_type:
            return c
    return None


def is_condition_true(va: VariantAutoscaling, condition_type: str) -> bool:
    c = get_condition(va, condition_type)
    return c is not None and c.status == "True"


def is_condition_false(va: VariantAutoscaling, condition_type: str) -> bool:
    c = get_condition(va, condition_type)
    return c is not None and c.status == "False"
