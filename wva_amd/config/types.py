"""System specification types — the solver's declarative JSON input.

Parity with /root/reference/pkg/config/types.go.  Dataclasses with JSON
round-trip helpers using the reference's field names (camelCase keys,
``slo-itl``/``slo-ttft``/``slo-tps`` SLO keys) so specs are interchangeable.

Unit conventions (the reference's de-facto contract, preserved exactly):
arrival rates are req/min in specs, req/s inside the sizing kernel, req/ms
inside the queue model; latencies are milliseconds everywhere.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field, fields, is_dataclass
from typing import Any, Dict, List, get_args, get_origin, get_type_hints


def _json_name(f) -> str:
    return f.metadata.get("json", f.name)


def _to_dict(obj: Any) -> Any:
    if is_dataclass(obj):
        out = {}
        for f in fields(obj):
            out[_json_name(f)] = _to_dict(getattr(obj, f.name))
        return out
    if isinstance(obj, list):
        return [_to_dict(x) for x in obj]
    if isinstance(obj, dict):
        return {k: _to_dict(v) for k, v in obj.items()}
    return obj


def _from_dict(cls, data: Any) -> Any:
    if data is None:
        return cls() if is_dataclass(cls) else None
    if is_dataclass(cls):
        hints = get_type_hints(cls)
        kwargs = {}
        for f in fields(cls):
            key = _json_name(f)
            if key in data:
                kwargs[f.name] = _convert(hints[f.name], data[key])
        return cls(**kwargs)
    return data


def _convert(tp, value):
    origin = get_origin(tp)
    if origin is list:
        (item_tp,) = get_args(tp)
        return [_convert(item_tp, v) for v in value or []]
    if origin is dict:
        _, val_tp = get_args(tp)
        return {k: _convert(val_tp, v) for k, v in (value or {}).items()}
    if is_dataclass(tp):
        return _from_dict(tp, value)
    if tp is float and value is not None:
        return float(value)
    if tp is int and value is not None:
        return int(value)
    return value


class SpecBase:
    """JSON round-trip mixin."""

    def to_dict(self) -> Dict[str, Any]:
        return _to_dict(self)

    def to_json(self, **kw) -> str:
        return json.dumps(self.to_dict(), **kw)

    @classmethod
    def from_dict(cls, data: Dict[str, Any]):
        return _from_dict(cls, data)

    @classmethod
    def from_json(cls, text: str):
        return cls.from_dict(json.loads(text))


def jfield(json_name: str, default=None, default_factory=None):
    if default_factory is not None:
        return field(default_factory=default_factory, metadata={"json": json_name})
    return field(default=default, metadata={"json": json_name})


# --------------------------------------------------------------- accelerators
@dataclass
class PowerSpec(SpecBase):
    """Accelerator power consumption points (Watts)."""

    idle: int = jfield("idle", 0)
    full: int = jfield("full", 0)
    mid_power: int = jfield("midPower", 0)
    mid_util: float = jfield("midUtil", 0.0)


@dataclass
class AcceleratorSpec(SpecBase):
    name: str = jfield("name", "")
    type: str = jfield("type", "")
    multiplicity: int = jfield("multiplicity", 1)  # cards per allocation unit
    mem_size: int = jfield("memSize", 0)  # GB
    mem_bw: int = jfield("memBW", 0)  # GB/s
    power: PowerSpec = jfield("power", default_factory=PowerSpec)
    cost: float = jfield("cost", 0.0)  # cents/hr


@dataclass
class AcceleratorData(SpecBase):
    spec: List[AcceleratorSpec] = jfield("accelerators", default_factory=list)


@dataclass
class AcceleratorCount(SpecBase):
    type: str = jfield("type", "")
    count: int = jfield("count", 0)


@dataclass
class CapacityData(SpecBase):
    count: List[AcceleratorCount] = jfield("count", default_factory=list)


# --------------------------------------------------------------------- models
@dataclass
class DecodeParmsSpec(SpecBase):
    """decode time = alpha + beta * batchSize (ms)."""

    alpha: float = jfield("alpha", 0.0)
    beta: float = jfield("beta", 0.0)


@dataclass
class PrefillParmsSpec(SpecBase):
    """prefill time = gamma + delta * inputTokens * batchSize (ms)."""

    gamma: float = jfield("gamma", 0.0)
    delta: float = jfield("delta", 0.0)


@dataclass
class ModelAcceleratorPerfData(SpecBase):
    name: str = jfield("name", "")  # model name
    acc: str = jfield("acc", "")  # accelerator name
    acc_count: int = jfield("accCount", 0)  # accelerator units used by model
    max_batch_size: int = jfield("maxBatchSize", 0)
    at_tokens: int = jfield("atTokens", 0)  # tokens/request assumed by maxBatchSize
    decode_parms: DecodeParmsSpec = jfield("decodeParms", default_factory=DecodeParmsSpec)
    prefill_parms: PrefillParmsSpec = jfield("prefillParms", default_factory=PrefillParmsSpec)


@dataclass
class ModelData(SpecBase):
    perf_data: List[ModelAcceleratorPerfData] = jfield("models", default_factory=list)


# -------------------------------------------------------------- service class
@dataclass
class ModelTarget(SpecBase):
    model: str = jfield("model", "")
    slo_itl: float = jfield("slo-itl", 0.0)  # ms
    slo_ttft: float = jfield("slo-ttft", 0.0)  # ms, including queueing
    slo_tps: float = jfield("slo-tps", 0.0)  # tokens/s


@dataclass
class ServiceClassSpec(SpecBase):
    name: str = jfield("name", "")
    priority: int = jfield("priority", 0)  # [1,100], lower value = higher prio
    model_targets: List[ModelTarget] = jfield("modelTargets", default_factory=list)


@dataclass
class ServiceClassData(SpecBase):
    spec: List[ServiceClassSpec] = jfield("serviceClasses", default_factory=list)


# -------------------------------------------------------------------- servers
@dataclass
class ServerLoadSpec(SpecBase):
    arrival_rate: float = jfield("arrivalRate", 0.0)  # req/min
    avg_in_tokens: int = jfield("avgInTokens", 0)
    avg_out_tokens: int = jfield("avgOutTokens", 0)


@dataclass
class AllocationData(SpecBase):
    accelerator: str = jfield("accelerator", "")
    num_replicas: int = jfield("numReplicas", 0)
    max_batch: int = jfield("maxBatch", 0)
    cost: float = jfield("cost", 0.0)
    itl_average: float = jfield("itlAverage", 0.0)
    ttft_average: float = jfield("ttftAverage", 0.0)
    load: ServerLoadSpec = jfield("load", default_factory=ServerLoadSpec)


@dataclass
class ServerSpec(SpecBase):
    name: str = jfield("name", "")
    class_name: str = jfield("class", "")
    model: str = jfield("model", "")
    keep_accelerator: bool = jfield("keepAccelerator", False)
    min_num_replicas: int = jfield("minNumReplicas", 0)
    max_batch_size: int = jfield("maxBatchSize", 0)  # override
    current_alloc: AllocationData = jfield("currentAlloc", default_factory=AllocationData)
    desired_alloc: AllocationData = jfield("desiredAlloc", default_factory=AllocationData)
    # MI355X extension (not in the reference spec JSON): per-server
    # service-time cs^2 for the mg1 analyzer; negative = unset -> the
    # global WVA_SERVICE_SCV applies (0 is a VALID measured value:
    # deterministic lengths).  Populated from the measured token
    # histogram in auto mode (collector.estimate_token_scv).
    service_scv: float = jfield("serviceSCV", -1.0)


@dataclass
class ServerData(SpecBase):
    spec: List[ServerSpec] = jfield("servers", default_factory=list)


@dataclass
class AllocationSolution(SpecBase):
    spec: Dict[str, AllocationData] = jfield("allocations", default_factory=dict)


# ------------------------------------------------------------------ optimizer
@dataclass
class OptimizerSpec(SpecBase):
    unlimited: bool = jfield("unlimited", False)
    delayed_best_effort: bool = jfield("delayedBestEffort", False)
    saturation_policy: str = jfield("saturationPolicy", "")
    # objective extension: "" / "cost" (default) or "cost+energy" — the
    # latter adds predicted power draw (accelerator power curve at the
    # allocation's utilization) priced at energyCostPerKWh cents/kWh
    objective: str = jfield("objective", "")
    energy_cost_per_kwh: float = jfield("energyCostPerKWh", 0.0)


@dataclass
class OptimizerData(SpecBase):
    spec: OptimizerSpec = jfield("optimizer", default_factory=OptimizerSpec)


# --------------------------------------------------------------------- system
@dataclass
class SystemSpec(SpecBase):
    accelerators: AcceleratorData = jfield("acceleratorData", default_factory=AcceleratorData)
    models: ModelData = jfield("modelData", default_factory=ModelData)
    service_classes: ServiceClassData = jfield("serviceClassData", default_factory=ServiceClassData)
    servers: ServerData = jfield("serverData", default_factory=ServerData)
    optimizer: OptimizerData = jfield("optimizerData", default_factory=OptimizerData)
    capacity: CapacityData = jfield("capacityData", default_factory=CapacityData)


@dataclass
class SystemData(SpecBase):
    spec: SystemSpec = jfield("system", default_factory=SystemSpec)
