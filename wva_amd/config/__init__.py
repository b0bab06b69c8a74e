"""Declarative system-spec types, defaults and policies (layer L4 input).

Parity with /root/reference/pkg/config/{types,defaults,config}.go, plus an
MI355X-first accelerator catalog (mi355x.py) replacing the reference's
NVIDIA-centric demo tables.
"""

from .types import (
    AcceleratorCount,
    AcceleratorData,
    AcceleratorSpec,
    AllocationData,
    AllocationSolution,
    CapacityData,
    DecodeParmsSpec,
    ModelAcceleratorPerfData,
    ModelData,
    ModelTarget,
    OptimizerData,
    OptimizerSpec,
    PowerSpec,
    PrefillParmsSpec,
    ServerData,
    ServerLoadSpec,
    ServerSpec,
    ServiceClassData,
    ServiceClassSpec,
    SystemData,
    SystemSpec,
)
from .defaults import (
    ACCEL_PENALTY_FACTOR,
    DEFAULT_HIGH_PRIORITY,
    DEFAULT_LOW_PRIORITY,
    DEFAULT_SERVICE_CLASS_NAME,
    DEFAULT_SERVICE_CLASS_PRIORITY,
    MAX_QUEUE_TO_BATCH_RATIO,
    SLO_MARGIN,
    SLO_PERCENTILE,
)
from .policies import SaturationPolicy
from .mi355x import MI355X_ACCELERATOR, MI355X_CATALOG, mi355x_accelerator_configmap

__all__ = [
    "AcceleratorCount",
    "AcceleratorData",
    "AcceleratorSpec",
    "AllocationData",
    "AllocationSolution",
    "CapacityData",
    "DecodeParmsSpec",
    "ModelAcceleratorPerfData",
    "ModelData",
    "ModelTarget",
    "OptimizerData",
    "OptimizerSpec",
    "PowerSpec",
    "PrefillParmsSpec",
    "ServerData",
    "ServerLoadSpec",
    "ServerSpec",
    "ServiceClassData",
    "ServiceClassSpec",
    "SystemData",
    "SystemSpec",
    "ACCEL_PENALTY_FACTOR",
    "DEFAULT_HIGH_PRIORITY",
    "DEFAULT_LOW_PRIORITY",
    "DEFAULT_SERVICE_CLASS_NAME",
    "DEFAULT_SERVICE_CLASS_PRIORITY",
    "MAX_QUEUE_TO_BATCH_RATIO",
    "SLO_MARGIN",
    "SLO_PERCENTILE",
    "SaturationPolicy",
    "MI355X_ACCELERATOR",
    "MI355X_CATALOG",
    "mi355x_accelerator_configmap",
]
