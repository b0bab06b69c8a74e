"""Actuator: emits current/desired/ratio gauges for HPA/KEDA.

Parity with /root/reference/internal/actuator/actuator.go — the actuator
does NOT scale Deployments itself; external HPA/KEDA consumes the gauges.
Replica source preference: Deployment status -> spec -> 1 (actuator.go:29-48).
"""

from __future__ import annotations

from ..api import v1alpha1
from ..kube import KubeClient
from .logger import log
from .metrics import MetricsEmitter
from .utils import get_deployment_with_backoff


class Actuator:
    def __init__(self, client: KubeClient) -> None:
        self.client = client
        self.metrics_emitter = MetricsEmitter()

    def _current_deployment_replicas(self, va: v1alpha1.VariantAutoscaling) -> int:
        deploy = get_deployment_with_backoff(self.client, va.name, va.namespace)
        if deploy.status.replicas >= 0:
            return deploy.status.replicas
        if deploy.spec.replicas is not None:
            return deploy.spec.replicas
        return 1

    def emit_metrics(self, va: v1alpha1.VariantAutoscaling) -> None:
        if va.status.desired_optimized_alloc.num_replicas < 0:
            log.info("Skipping EmitReplicaMetrics - desired replicas negative", variant=va.name)
            return
        try:
            current = self._current_deployment_replicas(va)
        except Exception as e:
            log.warn(
                "Could not get current deployment replicas, using VariantAutoscaling status",
                error=str(e),
                variant=va.name,
            )
            current = va.status.current_alloc.num_replicas
        try:
            self.metrics_emitter.emit_replica_metrics(
                va,
                current,
                va.status.desired_optimized_alloc.num_replicas,
                va.status.desired_optimized_alloc.accelerator,
            )
        except Exception as e:
            # metric emission failures must not break reconciliation
            log.error("Failed to emit optimization signals", variant=va.name, error=str(e))
            return
        log.info(
            "EmitReplicaMetrics completed",
            variant=va.name,
            current_replicas=current,
            desired_replicas=va.status.desired_optimized_alloc.num_replicas,
            accelerator=va.status.desired_optimized_alloc.accelerator,
        )
