"""ConfigMap -> SystemSpec adapters and retry plumbing.

Parity with /root/reference/internal/utils/utils.go: exponential-backoff
presets and retry wrappers (non-retryable NotFound/Invalid/Forbidden),
CreateSystemData (unlimited by default; unlike the reference, limited
mode is reachable via the controller ConfigMap — see create_system_data),
AddModelAcceleratorProfileToSystemData (string alpha/beta/gamma/delta
parsing), AddServerInfoToSystemData (KeepAccelerator:true pinning,
scale-to-zero via WVA_SCALE_TO_ZERO, maxBatchSize from the matching
profile), CreateOptimizedAlloc, FullName, FindModelSLO.
"""

from __future__ import annotations

import datetime
import math
import os
import random
import time
from dataclasses import dataclass
from typing import Callable, Dict, Optional, Tuple, TypeVar

import yaml

from ..api import v1alpha1
from ..config import (
    AcceleratorCount,
    AcceleratorSpec,
    AllocationData,
    AllocationSolution,
    DecodeParmsSpec,
    ModelAcceleratorPerfData,
    ModelTarget,
    OptimizerSpec,
    PrefillParmsSpec,
    ServerLoadSpec,
    ServerSpec,
    ServiceClassSpec,
    SystemData,
    SystemSpec,
)
from ..kube import ConfigMap, ConflictError, Deployment, KubeClient, KubeError
from .collector import ACCELERATOR_LABEL
from .interfaces import ServiceClassEntry, ServiceClassYaml
from .logger import log
from .promclient import PromAPI, PromQueryError

T = TypeVar("T")


# ------------------------------------------------------------------ backoff
@dataclass(frozen=True)
class Backoff:
    duration: float  # initial delay seconds
    factor: float
    jitter: float
    steps: int


STANDARD_BACKOFF = Backoff(duration=0.1, factor=2.0, jitter=0.1, steps=5)
RECONCILE_BACKOFF = Backoff(duration=0.5, factor=2.0, jitter=0.0, steps=5)
PROMETHEUS_BACKOFF = Backoff(duration=5.0, factor=2.0, jitter=0.1, steps=6)


def retry_with_backoff(fn: Callable[[], T], backoff: Backoff, what: str) -> T:
    """Run ``fn`` retrying transient failures with exponential backoff."""
    delay = backoff.duration
    last: Optional[Exception] = None
    for step in range(backoff.steps):
        try:
            return fn()
        except KubeError as e:
            if not e.retryable:
                raise
            last = e
        except PromQueryError as e:
            last = e
        log.error(f"transient error, retrying - {what}", attempt=step, error=str(last))
        if step < backoff.steps - 1:
            sleep = delay * (1.0 + backoff.jitter * random.random())
            time.sleep(sleep)
            delay *= backoff.factor
    raise last  # type: ignore[misc]


def get_deployment_with_backoff(client: KubeClient, name: str, namespace: str) -> Deployment:
    return retry_with_backoff(
        lambda: client.get(Deployment, name, namespace), STANDARD_BACKOFF, "Deployment"
    )


def get_configmap_with_backoff(client: KubeClient, name: str, namespace: str) -> ConfigMap:
    return retry_with_backoff(
        lambda: client.get(ConfigMap, name, namespace), STANDARD_BACKOFF, "ConfigMap"
    )


def get_variant_autoscaling_with_backoff(
    client: KubeClient, name: str, namespace: str
) -> v1alpha1.VariantAutoscaling:
    return retry_with_backoff(
        lambda: client.get(v1alpha1.VariantAutoscaling, name, namespace),
        STANDARD_BACKOFF,
        "VariantAutoscaling",
    )


def update_status_with_backoff(client: KubeClient, obj, what: str) -> None:
    """Status update with backoff; on 409 the object is re-read and the
    status re-applied before retrying (client-go's RetryOnConflict
    pattern — the reference retries the stale object and leaves conflict
    resolution to the next cycle, utils.go:91-104; resolving in-place
    keeps this cycle's optimization result instead of dropping it)."""
    state = {"obj": obj}

    def attempt():
        try:
            return client.update_status(state["obj"])
        except ConflictError:
            fresh = client.get(
                type(obj), obj.metadata.name, obj.metadata.namespace
            )
            fresh.status = obj.status
            state["obj"] = fresh
            raise

    retry_with_backoff(attempt, STANDARD_BACKOFF, what)


def validate_prometheus_api(prom: PromAPI, backoff: Backoff = PROMETHEUS_BACKOFF) -> None:
    """Query ``up`` with backoff; raises when Prometheus stays unreachable
    (startup is fatal on failure, controller.go:448-451)."""

    def probe():
        prom.query("up")
        return True

    retry_with_backoff(probe, backoff, "Prometheus API validation")


# ----------------------------------------------------------------- adapters
def check_value(x: float) -> bool:
    return not (math.isnan(x) or math.isinf(x))


def full_name(name: str, namespace: str) -> str:
    return f"{name}:{namespace}"


def create_system_data(
    accelerator_cm: Dict[str, Dict[str, str]],
    service_class_cm: Dict[str, str],
    optimizer_cm: Optional[Dict[str, str]] = None,
    inventory: Optional[Dict[str, Dict[str, object]]] = None,
) -> SystemData:
    """Adapter from ConfigMap payloads to the optimizer's SystemSpec.

    The reference hardwires Unlimited:true and leaves the greedy
    capacity-constrained solver dormant (utils.go:170-173).  Here limited
    mode is reachable: set ``WVA_OPTIMIZER_MODE: limited`` in the
    controller ConfigMap (plus optional ``WVA_SATURATION_POLICY`` and
    ``WVA_DELAYED_BEST_EFFORT``) and give the pool either statically —
    a ``capacity`` field per accelerator entry — or live, from the
    cluster's Node inventory (``WVA_INVENTORY: k8s``; ``inventory`` is
    collector.collect_inventory_k8s output keyed by product name, which
    takes precedence over the static field)."""
    sd = SystemData(spec=SystemSpec())
    optimizer_cm = optimizer_cm or {}
    inventory = inventory or {}

    for key, val in accelerator_cm.items():
        try:
            cost = float(val["cost"])
        except (KeyError, ValueError):
            log.warn("failed to parse accelerator cost in configmap, skipping accelerator", name=key)
            continue
        spec = AcceleratorSpec(
            name=key,
            type=val.get("device", ""),
            multiplicity=1,
            cost=cost,
        )
        # MI355X extension: optional memory/bandwidth fields for KV sizing
        try:
            spec.mem_size = int(val.get("memSize", 0))
            spec.mem_bw = int(val.get("memBW", 0))
        except ValueError:
            pass
        sd.spec.accelerators.spec.append(spec)
        if key in inventory:
            # live node inventory wins over the static capacity field
            sd.spec.capacity.count.append(
                AcceleratorCount(type=spec.type, count=int(inventory[key]["count"]))
            )
        elif "capacity" in val:
            try:
                sd.spec.capacity.count.append(
                    AcceleratorCount(type=spec.type, count=int(val["capacity"]))
                )
            except ValueError:
                log.warn("failed to parse accelerator capacity, ignoring", name=key)

    for key, val in service_class_cm.items():
        sc = parse_service_class_yaml(key, val)
        if sc is None:
            continue
        sd.spec.service_classes.spec.append(
            ServiceClassSpec(
                name=sc.name,
                priority=sc.priority,
                model_targets=[
                    ModelTarget(model=e.model, slo_itl=float(e.slo_tpot), slo_ttft=float(e.slo_ttft))
                    for e in sc.data
                ],
            )
        )

    mode = optimizer_cm.get("WVA_OPTIMIZER_MODE", "unlimited").lower()
    try:
        energy_price = float(optimizer_cm.get("WVA_ENERGY_COST_PER_KWH", "0"))
    except ValueError:
        energy_price = 0.0
    sd.spec.optimizer.spec = OptimizerSpec(
        unlimited=mode != "limited",
        delayed_best_effort=optimizer_cm.get("WVA_DELAYED_BEST_EFFORT", "").lower() == "true",
        saturation_policy=optimizer_cm.get("WVA_SATURATION_POLICY", ""),
        objective=optimizer_cm.get("WVA_OBJECTIVE", ""),
        energy_cost_per_kwh=energy_price,
    )
    return sd


def parse_service_class_yaml(key: str, text: str) -> Optional[ServiceClassYaml]:
    try:
        raw = yaml.safe_load(text)
        if not isinstance(raw, dict):
            raise ValueError("not a mapping")
        entries = [
            ServiceClassEntry(
                model=d.get("model", ""),
                slo_tpot=int(d.get("slo-tpot", 0)),
                slo_ttft=int(d.get("slo-ttft", 0)),
            )
            for d in raw.get("data", []) or []
        ]
        return ServiceClassYaml(
            name=raw.get("name", ""), priority=int(raw.get("priority", 0)), data=entries
        )
    except Exception as e:
        log.warn("failed to parse service class data, skipping service class", key=key, err=str(e))
        return None


def add_model_accelerator_profile_to_system_data(
    sd: SystemData, model_name: str, profile: v1alpha1.AcceleratorProfile
) -> None:
    """Parse the profile's string alpha/beta/gamma/delta into perf data;
    raises ValueError on malformed input (utils.go:185-234)."""
    decode = profile.perf_parms.decode_parms
    if len(decode) < 2:
        raise ValueError("length of decodeParms should be 2")
    alpha = float(decode["alpha"])
    beta = float(decode["beta"])
    prefill = profile.perf_parms.prefill_parms
    if len(prefill) < 2:
        raise ValueError("length of prefillParms should be 2")
    gamma = float(prefill["gamma"])
    delta = float(prefill["delta"])

    sd.spec.models.perf_data.append(
        ModelAcceleratorPerfData(
            name=model_name,
            acc=profile.acc,
            acc_count=profile.acc_count,
            max_batch_size=profile.max_batch_size,
            decode_parms=DecodeParmsSpec(alpha=alpha, beta=beta),
            prefill_parms=PrefillParmsSpec(gamma=gamma, delta=delta),
        )
    )


def scale_to_zero_enabled() -> bool:
    return os.environ.get("WVA_SCALE_TO_ZERO", "").lower() == "true"


def add_server_info_to_system_data(
    sd: SystemData, va: v1alpha1.VariantAutoscaling, class_name: str
) -> None:
    """status.currentAlloc -> ServerSpec (utils.go:237-311): keepAccelerator
    pinned true, minNumReplicas 0 iff scale-to-zero, maxBatchSize from the
    profile matching the accelerator label."""

    def parse(s: str) -> float:
        try:
            v = float(s)
        except (TypeError, ValueError):
            return 0.0
        return v if check_value(v) else 0.0

    cur = va.status.current_alloc
    load = ServerLoadSpec(
        arrival_rate=parse(cur.load.arrival_rate),
        avg_in_tokens=int(parse(cur.load.avg_input_tokens)),
        avg_out_tokens=int(parse(cur.load.avg_output_tokens)),
    )
    alloc = AllocationData(
        accelerator=cur.accelerator,
        num_replicas=cur.num_replicas,
        max_batch=cur.max_batch,
        cost=parse(cur.variant_cost),
        itl_average=parse(cur.itl_average),
        ttft_average=parse(cur.ttft_average),
        load=load,
    )
    server = ServerSpec(
        name=full_name(va.name, va.namespace),
        class_name=class_name,
        model=va.spec.model_id,
        keep_accelerator=True,
        min_num_replicas=0 if scale_to_zero_enabled() else 1,
        current_alloc=alloc,
        desired_alloc=AllocationData(),
    )
    acc_name = va.metadata.labels.get(ACCELERATOR_LABEL, "")
    for ap in va.spec.model_profile.accelerators:
        if ap.acc == acc_name:
            if ap.max_batch_size > 0:
                server.max_batch_size = ap.max_batch_size
            break
    sd.spec.servers.spec.append(server)


def create_optimized_alloc(
    name: str, namespace: str, solution: AllocationSolution
) -> v1alpha1.OptimizedAlloc:
    server_name = full_name(name, namespace)
    if server_name not in solution.spec:
        raise KeyError(f"server {server_name} not found")
    data = solution.spec[server_name]
    return v1alpha1.OptimizedAlloc(
        lastRunTime=datetime.datetime.now(datetime.timezone.utc),
        accelerator=data.accelerator,
        numReplicas=data.num_replicas,
    )


def find_model_slo(
    cm_data: Dict[str, str], target_model: str
) -> Tuple[ServiceClassEntry, str]:
    """Scan service-class YAMLs for the model; returns (entry, class name)."""
    for key, val in cm_data.items():
        sc = parse_service_class_yaml(key, val)
        if sc is None:
            raise ValueError(f"failed to parse {key}")
        for entry in sc.data:
            if entry.model == target_model:
                return entry, sc.name
    raise KeyError(f"model {target_model!r} not found in any service class")
