"""Typed Kubernetes objects the controller touches: ConfigMap,
Deployment, and the coordination.k8s.io Lease used for leader election."""

from __future__ import annotations

import datetime
from typing import Dict, Optional

from pydantic import BaseModel, ConfigDict, Field

from ..api.v1alpha1.types import ObjectMeta


class _Base(BaseModel):
    model_config = ConfigDict(populate_by_name=True)


class ConfigMap(_Base):
    api_version: str = Field(alias="apiVersion", default="v1")
    kind: str = "ConfigMap"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    data: Dict[str, str] = Field(default_factory=dict)

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace


class DeploymentSpec(_Base):
    replicas: Optional[int] = None


class DeploymentStatus(_Base):
    replicas: int = 0
    ready_replicas: int = Field(alias="readyReplicas", default=0)


class LeaseSpec(_Base):
    holder_identity: Optional[str] = Field(alias="holderIdentity", default=None)
    lease_duration_seconds: Optional[int] = Field(alias="leaseDurationSeconds", default=None)
    acquire_time: Optional[datetime.datetime] = Field(alias="acquireTime", default=None)
    renew_time: Optional[datetime.datetime] = Field(alias="renewTime", default=None)
    lease_transitions: int = Field(alias="leaseTransitions", default=0)


class Lease(_Base):
    """coordination.k8s.io/v1 Lease — leader-election primitive (the
    reference's manager elects through the same object via
    controller-runtime, LeaderElectionID 72dd1cf1.llm-d.ai)."""

    api_version: str = Field(alias="apiVersion", default="coordination.k8s.io/v1")
    kind: str = "Lease"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: LeaseSpec = Field(default_factory=LeaseSpec)

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace


class Deployment(_Base):
    api_version: str = Field(alias="apiVersion", default="apps/v1")
    kind: str = "Deployment"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: DeploymentSpec = Field(default_factory=DeploymentSpec)
    status: DeploymentStatus = Field(default_factory=DeploymentStatus)

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace


class NodeStatus(_Base):
    capacity: Dict[str, str] = Field(default_factory=dict)
    allocatable: Dict[str, str] = Field(default_factory=dict)


class Node(_Base):
    """core/v1 Node (cluster-scoped) — carries the GPU inventory the
    limited-mode solver consumes: ``amd.com/gpu`` extended-resource
    counts in status.allocatable plus the ``amd.com/gpu.product`` label
    (the convention deploy/kind-emulator/setup.sh and
    deploy/emulator/amd-gpu-node-labels.yaml install).  The reference
    stubs this inventory (collector.go:37-42); here it is live."""

    api_version: str = Field(alias="apiVersion", default="v1")
    kind: str = "Node"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    status: NodeStatus = Field(default_factory=NodeStatus)

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace
