"""Typed Kubernetes objects the controller touches: ConfigMap, Deployment."""

from __future__ import annotations

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
