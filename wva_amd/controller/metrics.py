"""Custom metrics emitter.

Parity with /root/reference/internal/metrics/metrics.go: counter
``inferno_replica_scaling_total`` and gauges ``inferno_desired_replicas`` /
``inferno_current_replicas`` / ``inferno_desired_ratio`` with labels
(variant_name, namespace, accelerator_type), including the 0->N scale-up
encoding ratio = N (metrics.go:118-124).

Additions over the reference:
- ``wva_solver_duration_seconds`` histogram — the solve wall-clock promoted
  to a first-class metric (it is the BASELINE headline metric; the
  reference only prints it via String()).
"""

from __future__ import annotations

from typing import Optional

from prometheus_client import CollectorRegistry, Counter, Gauge, Histogram

from . import constants

_replica_scaling_total: Optional[Counter] = None
_desired_replicas: Optional[Gauge] = None
_current_replicas: Optional[Gauge] = None
_desired_ratio: Optional[Gauge] = None
_solver_duration: Optional[Histogram] = None
_cycle_phase_duration: Optional[Histogram] = None


def init_metrics(registry: CollectorRegistry) -> None:
    """Register all custom metrics with the provided registry and hook the
    solver's duration observer."""
    from ..solver.optimizer import register_solve_observer

    register_solve_observer(observe_solver_duration)
    global _replica_scaling_total, _desired_replicas, _current_replicas, _desired_ratio, _solver_duration
    _replica_scaling_total = Counter(
        constants.INFERNO_REPLICA_SCALING_TOTAL,
        "Total number of replica scaling operations",
        [
            constants.LABEL_VARIANT_NAME,
            constants.LABEL_NAMESPACE,
            constants.LABEL_DIRECTION,
            constants.LABEL_REASON,
        ],
        registry=registry,
    )
    _desired_replicas = Gauge(
        constants.INFERNO_DESIRED_REPLICAS,
        "Desired number of replicas for each variant",
        [constants.LABEL_VARIANT_NAME, constants.LABEL_NAMESPACE, constants.LABEL_ACCELERATOR_TYPE],
        registry=registry,
    )
    _current_replicas = Gauge(
        constants.INFERNO_CURRENT_REPLICAS,
        "Current number of replicas for each variant",
        [constants.LABEL_VARIANT_NAME, constants.LABEL_NAMESPACE, constants.LABEL_ACCELERATOR_TYPE],
        registry=registry,
    )
    _desired_ratio = Gauge(
        constants.INFERNO_DESIRED_RATIO,
        "Ratio of desired to current replicas for each variant",
        [constants.LABEL_VARIANT_NAME, constants.LABEL_NAMESPACE, constants.LABEL_ACCELERATOR_TYPE],
        registry=registry,
    )
    _solver_duration = Histogram(
        constants.WVA_SOLVER_DURATION_SECONDS,
        "Wall-clock duration of one global optimization solve",
        registry=registry,
        buckets=(
            1e-5, 2.5e-5, 5e-5, 1e-4, 2.5e-4, 5e-4, 1e-3, 2.5e-3, 5e-3,
            1e-2, 2.5e-2, 5e-2, 0.1, 0.25, 0.5, 1.0, 2.5,
        ),
    )
    global _cycle_phase_duration
    _cycle_phase_duration = Histogram(
        constants.WVA_CYCLE_PHASE_DURATION_SECONDS,
        "Wall-clock duration of each reconcile-cycle phase",
        ["phase"],  # config / prepare / analyze / optimize / apply
        registry=registry,
        buckets=(
            1e-4, 5e-4, 1e-3, 2.5e-3, 5e-3, 1e-2, 2.5e-2, 5e-2,
            0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0,
        ),
    )


def reset_metrics() -> None:
    """Drop metric handles (tests use fresh registries)."""
    global _replica_scaling_total, _desired_replicas, _current_replicas, _desired_ratio
    global _solver_duration, _cycle_phase_duration
    _replica_scaling_total = _desired_replicas = _current_replicas = _desired_ratio = None
    _solver_duration = _cycle_phase_duration = None


def observe_solver_duration(seconds: float) -> None:
    if _solver_duration is not None:
        _solver_duration.observe(seconds)


def observe_cycle_phase(phase: str, seconds: float) -> None:
    if _cycle_phase_duration is not None:
        _cycle_phase_duration.labels(phase).observe(seconds)


class MetricsEmitter:
    """Emission of replica metrics for external autoscalers (HPA/KEDA)."""

    def emit_replica_scaling_metrics(self, va, direction: str, reason: str) -> None:
        if _replica_scaling_total is None:
            raise RuntimeError("replicaScalingTotal metric not initialized")
        _replica_scaling_total.labels(va.name, va.namespace, direction, reason).inc()

    def emit_replica_metrics(
        self, va, current: int, desired: int, accelerator_type: str
    ) -> None:
        if _current_replicas is None or _desired_replicas is None or _desired_ratio is None:
            raise RuntimeError("replica metrics not initialized")
        labels = (va.name, va.namespace, accelerator_type)
        _current_replicas.labels(*labels).set(float(current))
        _desired_replicas.labels(*labels).set(float(desired))
        # 0 -> N scale-up is encoded as ratio = N
        if current == 0:
            _desired_ratio.labels(*labels).set(float(desired))
        else:
            _desired_ratio.labels(*labels).set(float(desired) / float(current))
