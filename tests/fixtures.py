"""Shared test fixtures: an MI355X-first system mirroring the reference's
greedy-test fixture structure (/root/reference/pkg/solver/greedy_test.go:13-100
built A100/H100 + llama-7b/13b; here the pool is MI355X/MI300X/L40S with
llama-8b/llama-70b and Premium/Freemium classes)."""

from wva_amd.config import (
    AcceleratorCount,
    AcceleratorData,
    AcceleratorSpec,
    AllocationData,
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
    SystemSpec,
)
from wva_amd.core import System

MI355X = AcceleratorSpec(
    name="MI355X",
    type="AMD-MI355X-288GB",
    multiplicity=1,
    mem_size=288,
    mem_bw=8000,
    power=PowerSpec(idle=140, full=1400, mid_power=900, mid_util=0.6),
    cost=85.0,
)
MI300X = AcceleratorSpec(
    name="MI300X",
    type="AMD-MI300X-192GB",
    multiplicity=1,
    mem_size=192,
    mem_bw=5300,
    power=PowerSpec(idle=130, full=750, mid_power=520, mid_util=0.6),
    cost=65.0,
)
L40S = AcceleratorSpec(
    name="L40S",
    type="EMU-L40S-48GB",
    multiplicity=1,
    mem_size=48,
    mem_bw=864,
    power=PowerSpec(idle=30, full=350, mid_power=240, mid_util=0.6),
    cost=23.0,
)


def perf(model, acc, acc_count=1, max_batch=256, alpha=6.0, beta=0.05, gamma=10.0, delta=0.05):
    return ModelAcceleratorPerfData(
        name=model,
        acc=acc,
        acc_count=acc_count,
        max_batch_size=max_batch,
        decode_parms=DecodeParmsSpec(alpha=alpha, beta=beta),
        prefill_parms=PrefillParmsSpec(gamma=gamma, delta=delta),
    )


def server_spec(
    name,
    model="llama-8b",
    class_name="Premium",
    arrival_rate=60.0,  # req/min
    in_tokens=128,
    out_tokens=128,
    min_replicas=1,
    max_batch=8,
    keep_accelerator=False,
    cur_accelerator="",
    cur_replicas=0,
):
    return ServerSpec(
        name=name,
        class_name=class_name,
        model=model,
        keep_accelerator=keep_accelerator,
        min_num_replicas=min_replicas,
        max_batch_size=max_batch,
        current_alloc=AllocationData(
            accelerator=cur_accelerator,
            num_replicas=cur_replicas,
            load=ServerLoadSpec(
                arrival_rate=arrival_rate,
                avg_in_tokens=in_tokens,
                avg_out_tokens=out_tokens,
            ),
        ),
    )


def make_spec(servers=None, unlimited=True, capacity=None, saturation_policy="", delayed_best_effort=False):
    import copy

    servers = servers if servers is not None else [server_spec("s1:default")]
    capacity = capacity or []
    return SystemSpec(
        accelerators=AcceleratorData(spec=copy.deepcopy([MI355X, MI300X, L40S])),
        models=ModelData(
            perf_data=[
                perf("llama-8b", "MI355X", alpha=4.0, beta=0.03, gamma=8.0, delta=0.03),
                perf("llama-8b", "MI300X", alpha=7.0, beta=0.05, gamma=12.0, delta=0.05),
                perf("llama-8b", "L40S", alpha=20.0, beta=0.6, gamma=60.0, delta=0.4),
                perf("llama-70b", "MI355X", acc_count=4, alpha=9.0, beta=0.06, gamma=18.0, delta=0.06),
                perf("llama-70b", "MI300X", acc_count=8, alpha=15.0, beta=0.10, gamma=30.0, delta=0.10),
            ]
        ),
        service_classes=ServiceClassData(
            spec=[
                ServiceClassSpec(
                    name="Premium",
                    priority=1,
                    model_targets=[
                        ModelTarget(model="llama-8b", slo_itl=20.0, slo_ttft=2000.0),
                        ModelTarget(model="llama-70b", slo_itl=30.0, slo_ttft=4000.0),
                    ],
                ),
                ServiceClassSpec(
                    name="Freemium",
                    priority=10,
                    model_targets=[
                        ModelTarget(model="llama-8b", slo_itl=60.0, slo_ttft=8000.0),
                        ModelTarget(model="llama-70b", slo_itl=90.0, slo_ttft=10000.0),
                    ],
                ),
            ]
        ),
        servers=ServerData(spec=servers),
        optimizer=OptimizerData(
            spec=OptimizerSpec(
                unlimited=unlimited,
                delayed_best_effort=delayed_best_effort,
                saturation_policy=saturation_policy,
            )
        ),
        capacity=CapacityData(count=[AcceleratorCount(type=t, count=c) for t, c in capacity]),
    )


def make_system(**kw):
    spec = make_spec(**kw)
    system = System()
    opt_spec = system.set_from_spec(spec)
    return system, opt_spec
