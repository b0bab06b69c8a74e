"""Controller-test fixtures: fake cluster payloads mirroring
/root/reference/test/utils/unitutils.go (ConfigMaps, VAs, MockPromAPI
wiring) with an MI355X-first accelerator table."""

import json

from wva_amd.api import v1alpha1
from wva_amd.api.v1alpha1.types import ObjectMeta
from wva_amd.controller import collector
from wva_amd.controller.reconciler import (
    ACCELERATOR_COSTS_CM,
    CONFIG_MAP_NAME,
    CONFIG_MAP_NAMESPACE,
    SERVICE_CLASSES_CM,
)
from wva_amd.kube import ConfigMap, Deployment, DeploymentSpec, DeploymentStatus, InMemoryKubeClient

PREMIUM_YAML = """\
name: Premium
priority: 1
data:
  - model: default/llama-8b
    slo-tpot: 24
    slo-ttft: 500
  - model: llama-8b
    slo-tpot: 24
    slo-ttft: 500
"""

FREEMIUM_YAML = """\
name: Freemium
priority: 10
data:
  - model: default/llama-70b
    slo-tpot: 200
    slo-ttft: 2000
"""


def accelerator_unit_costs() -> dict:
    return {
        "MI355X": json.dumps({"device": "AMD-MI355X-288GB", "cost": "85.00", "memSize": "288", "memBW": "8000"}),
        "MI300X": json.dumps({"device": "AMD-MI300X-192GB", "cost": "65.00", "memSize": "192", "memBW": "5300"}),
        "L40S": json.dumps({"device": "EMU-L40S-48GB", "cost": "23.00", "memSize": "48", "memBW": "864"}),
    }


def make_cluster(opt_interval="1s"):
    """Fake cluster with the three ConfigMaps installed."""
    client = InMemoryKubeClient()
    client.create(
        ConfigMap(
            metadata=ObjectMeta(name=CONFIG_MAP_NAME, namespace=CONFIG_MAP_NAMESPACE),
            data={"GLOBAL_OPT_INTERVAL": opt_interval, "PROMETHEUS_BASE_URL": "https://prom.test:9090"},
        )
    )
    client.create(
        ConfigMap(
            metadata=ObjectMeta(name=ACCELERATOR_COSTS_CM, namespace=CONFIG_MAP_NAMESPACE),
            data=accelerator_unit_costs(),
        )
    )
    client.create(
        ConfigMap(
            metadata=ObjectMeta(name=SERVICE_CLASSES_CM, namespace=CONFIG_MAP_NAMESPACE),
            data={"premium.yaml": PREMIUM_YAML, "freemium.yaml": FREEMIUM_YAML},
        )
    )
    return client


def make_deployment(client, name="vllm-llama", namespace="default", replicas=2):
    return client.create(
        Deployment(
            metadata=ObjectMeta(name=name, namespace=namespace),
            spec=DeploymentSpec(replicas=replicas),
            status=DeploymentStatus(replicas=replicas, readyReplicas=replicas),
        )
    )


def make_va(
    client,
    name="vllm-llama",
    namespace="default",
    model_id="default/llama-8b",
    accelerator="MI355X",
    max_batch=256,
    alpha="6.958",
    beta="0.042",
    gamma="20.0",
    delta="0.1",
    acc_count=1,
):
    va = v1alpha1.VariantAutoscaling(
        metadata=ObjectMeta(
            name=name,
            namespace=namespace,
            labels={"inference.optimization/acceleratorName": accelerator},
        ),
        spec=v1alpha1.VariantAutoscalingSpec(
            modelID=model_id,
            sloClassRef=v1alpha1.ConfigMapKeyRef(name=SERVICE_CLASSES_CM, key="premium.yaml"),
            modelProfile=v1alpha1.ModelProfile(
                accelerators=[
                    v1alpha1.AcceleratorProfile(
                        acc=accelerator,
                        accCount=acc_count,
                        maxBatchSize=max_batch,
                        perfParms=v1alpha1.PerfParms(
                            decodeParms={"alpha": alpha, "beta": beta},
                            prefillParms={"gamma": gamma, "delta": delta},
                        ),
                    )
                ]
            ),
        ),
    )
    return client.create(va)


def set_load_metrics(
    prom, model, namespace, arrival_rps=2.0, in_tokens=128.0, out_tokens=128.0, ttft_s=0.1, itl_s=0.01
):
    """Wire the five collector queries on a MockPromAPI."""
    prom.set_result(collector.arrival_query(model, namespace), arrival_rps)
    prom.set_result(collector.avg_prompt_tokens_query(model, namespace), in_tokens)
    prom.set_result(collector.avg_generation_tokens_query(model, namespace), out_tokens)
    prom.set_result(collector.ttft_query(model, namespace), ttft_s)
    prom.set_result(collector.itl_query(model, namespace), itl_s)
