"""Controller-internal interface types.

Parity with /root/reference/internal/interfaces/types.go: the analyzer
response shapes, the service-class ConfigMap YAML shapes (``slo-tpot`` /
``slo-ttft`` keys), and the Prometheus client configuration.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List

from ..core import Allocation


@dataclass
class ModelAcceleratorAllocation:
    """Allocation details of one accelerator for a variant."""

    allocation: Allocation
    required_prefill_qps: float = 0.0
    required_decode_qps: float = 0.0
    reason: str = ""


@dataclass
class ModelAnalyzeResponse:
    """Feasible allocations for all accelerators (acc name -> allocation)."""

    allocations: Dict[str, ModelAcceleratorAllocation] = field(default_factory=dict)


@dataclass
class ServiceClassEntry:
    model: str
    slo_tpot: int = 0  # ms
    slo_ttft: int = 0  # ms


@dataclass
class ServiceClassYaml:
    name: str
    priority: int
    data: List[ServiceClassEntry] = field(default_factory=list)


@dataclass
class PrometheusConfig:
    """Prometheus client configuration; HTTPS-only."""

    base_url: str = ""
    insecure_skip_verify: bool = False
    ca_cert_path: str = ""
    client_cert_path: str = ""
    client_key_path: str = ""
    server_name: str = ""
    bearer_token: str = ""
    token_path: str = ""
