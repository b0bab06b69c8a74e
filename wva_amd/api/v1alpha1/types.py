"""VariantAutoscaling CRD schema (group llmd.ai, version v1alpha1).

Pydantic models matching the reference CRD byte-for-byte at the JSON level
(/root/reference/api/v1alpha1/variantautoscaling_types.go): camelCase field
names, numeric status fields as strings with pattern ``^\\d+(\\.\\d+)?$``
(types.go:107-116), printcolumns, shortname ``va``.
"""

from __future__ import annotations

import datetime
from typing import Dict, List, Optional

from pydantic import BaseModel, ConfigDict, Field

GROUP = "llmd.ai"
VERSION = "v1alpha1"
KIND = "VariantAutoscaling"
SHORT_NAME = "va"

NUMERIC_STRING = r"^\d+(\.\d+)?$"

# kubectl printcolumns (variantautoscaling_types.go:160-165)
PRINT_COLUMNS = [
    ("Model", ".spec.modelID"),
    ("Accelerator", ".status.currentAlloc.accelerator"),
    ("CurrentReplicas", ".status.currentAlloc.numReplicas"),
    ("Optimized", ".status.desiredOptimizedAlloc.numReplicas"),
    ("MetricsReady", ".status.conditions[?(@.type=='MetricsAvailable')].status"),
    ("Age", ".metadata.creationTimestamp"),
]


class _Base(BaseModel):
    model_config = ConfigDict(populate_by_name=True, validate_assignment=False)

    def to_dict(self) -> dict:
        return self.model_dump(by_alias=True, exclude_none=True)


class ConfigMapKeyRef(_Base):
    name: str = Field(min_length=1)
    key: str = Field(min_length=1)


class PerfParms(_Base):
    """decodeParms keys alpha/beta: itl = alpha + beta * batch;
    prefillParms keys gamma/delta: ttft = gamma + delta * tokens * batch."""

    decode_parms: Dict[str, str] = Field(alias="decodeParms", default_factory=dict)
    prefill_parms: Dict[str, str] = Field(alias="prefillParms", default_factory=dict)


class AcceleratorProfile(_Base):
    acc: str = Field(min_length=1)
    acc_count: int = Field(alias="accCount", ge=1)
    perf_parms: PerfParms = Field(alias="perfParms", default_factory=PerfParms)
    max_batch_size: int = Field(alias="maxBatchSize", ge=1)


class ModelProfile(_Base):
    accelerators: List[AcceleratorProfile] = Field(min_length=1)


class VariantAutoscalingSpec(_Base):
    # modelID minLength=1 is enforced by the CRD schema (deploy/crd/);
    # the reconciler skips empty modelIDs at runtime like the reference
    model_id: str = Field(alias="modelID", default="")
    slo_class_ref: ConfigMapKeyRef = Field(
        alias="sloClassRef", default_factory=lambda: ConfigMapKeyRef(name="x", key="x")
    )
    model_profile: ModelProfile = Field(
        alias="modelProfile",
        default_factory=lambda: ModelProfile(
            accelerators=[AcceleratorProfile(acc="MI355X", accCount=1, maxBatchSize=1)]
        ),
    )


class LoadProfile(_Base):
    arrival_rate: str = Field(alias="arrivalRate", default="0")
    avg_input_tokens: str = Field(alias="avgInputTokens", default="0")
    avg_output_tokens: str = Field(alias="avgOutputTokens", default="0")


class Allocation(_Base):
    accelerator: str = ""
    num_replicas: int = Field(alias="numReplicas", default=0, ge=0)
    max_batch: int = Field(alias="maxBatch", default=0, ge=0)
    variant_cost: str = Field(alias="variantCost", pattern=NUMERIC_STRING, default="0")
    itl_average: str = Field(alias="itlAverage", pattern=NUMERIC_STRING, default="0")
    ttft_average: str = Field(alias="ttftAverage", pattern=NUMERIC_STRING, default="0")
    load: LoadProfile = Field(default_factory=LoadProfile)


class OptimizedAlloc(_Base):
    last_run_time: Optional[datetime.datetime] = Field(alias="lastRunTime", default=None)
    accelerator: str = ""
    num_replicas: int = Field(alias="numReplicas", default=0, ge=0)


class ActuationStatus(_Base):
    applied: bool = False


class Condition(_Base):
    type: str
    status: str  # "True" | "False" | "Unknown"
    observed_generation: int = Field(alias="observedGeneration", default=0)
    last_transition_time: Optional[datetime.datetime] = Field(
        alias="lastTransitionTime", default=None
    )
    reason: str = ""
    message: str = ""


class VariantAutoscalingStatus(_Base):
    current_alloc: Allocation = Field(alias="currentAlloc", default_factory=Allocation)
    desired_optimized_alloc: OptimizedAlloc = Field(
        alias="desiredOptimizedAlloc", default_factory=OptimizedAlloc
    )
    actuation: ActuationStatus = Field(default_factory=ActuationStatus)
    conditions: List[Condition] = Field(default_factory=list)


class OwnerReference(_Base):
    api_version: str = Field(alias="apiVersion", default="apps/v1")
    kind: str = "Deployment"
    name: str = ""
    uid: str = ""
    controller: bool = False
    block_owner_deletion: bool = Field(alias="blockOwnerDeletion", default=False)


class ObjectMeta(_Base):
    name: str = ""
    namespace: str = "default"
    labels: Dict[str, str] = Field(default_factory=dict)
    annotations: Dict[str, str] = Field(default_factory=dict)
    generation: int = 1
    resource_version: int = Field(alias="resourceVersion", default=0)
    uid: str = ""
    creation_timestamp: Optional[datetime.datetime] = Field(
        alias="creationTimestamp", default=None
    )
    deletion_timestamp: Optional[datetime.datetime] = Field(
        alias="deletionTimestamp", default=None
    )
    owner_references: List[OwnerReference] = Field(
        alias="ownerReferences", default_factory=list
    )


class VariantAutoscaling(_Base):
    api_version: str = Field(alias="apiVersion", default=f"{GROUP}/{VERSION}")
    kind: str = KIND
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: VariantAutoscalingSpec = Field(default_factory=VariantAutoscalingSpec)
    status: VariantAutoscalingStatus = Field(default_factory=VariantAutoscalingStatus)

    @property
    def name(self) -> str:
        return self.metadata.name

    @property
    def namespace(self) -> str:
        return self.metadata.namespace


class VariantAutoscalingList(_Base):
    api_version: str = Field(alias="apiVersion", default=f"{GROUP}/{VERSION}")
    kind: str = "VariantAutoscalingList"
    items: List[VariantAutoscaling] = Field(default_factory=list)
