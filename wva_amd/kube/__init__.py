"""Minimal Kubernetes client abstraction.

The reference is a kubebuilder controller talking to a real API server via
controller-runtime.  This rebuild targets the same protocol surface (typed
get/list/patch/status-update with NotFound/Invalid/Forbidden error classes,
ownerReference garbage collection, resourceVersion bumping) behind a small
``KubeClient`` protocol with two implementations:

- :class:`InMemoryKubeClient` — the envtest analog used by the component
  and e2e test tiers;
- :class:`HTTPKubeClient` — the real-cluster client speaking the API
  server's REST conventions (in-cluster service-account config or
  explicit endpoint); the controller only sees the protocol.
"""

from .errors import ConflictError, ForbiddenError, GoneError, InvalidError, KubeError, NotFoundError
from .objects import ConfigMap, Deployment, DeploymentSpec, DeploymentStatus, Lease, LeaseSpec, Node, NodeStatus
from .client import InMemoryKubeClient, KubeClient
from .http_client import HTTPKubeClient

__all__ = [
    "KubeError",
    "NotFoundError",
    "InvalidError",
    "ForbiddenError",
    "ConflictError",
    "GoneError",
    "ConfigMap",
    "Deployment",
    "Lease",
    "LeaseSpec",
    "Node",
    "NodeStatus",
    "DeploymentSpec",
    "DeploymentStatus",
    "KubeClient",
    "InMemoryKubeClient",
    "HTTPKubeClient",
]
