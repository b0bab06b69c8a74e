"""Live GPU inventory from cluster Nodes (the reference stubs this:
collector.go:37-42 / PContext-skipped collector suite).  Here
`collect_inventory_k8s` reads `amd.com/gpu` allocatable + product
labels off real Node objects, and limited mode turns the result into
the solver's capacity pool (`WVA_INVENTORY: k8s`)."""

import pytest
from prometheus_client import CollectorRegistry

from wva_amd.api import v1alpha1
from wva_amd.api.v1alpha1.types import ObjectMeta
from wva_amd.controller import collector
from wva_amd.controller import metrics as ctrl_metrics
from wva_amd.controller.promclient import MockPromAPI
from wva_amd.controller.reconciler import (
    CONFIG_MAP_NAME,
    CONFIG_MAP_NAMESPACE,
    VariantAutoscalingReconciler,
)
from wva_amd.controller.utils import create_system_data
from wva_amd.kube import ConfigMap, InMemoryKubeClient, Node, NodeStatus
from kube_fixtures import (
    accelerator_unit_costs,
    make_cluster,
    make_deployment,
    make_va,
    set_load_metrics,
)


def make_node(client, name, product="MI355X", gpus=8, vendor="amd.com",
              labeled=True, allocatable=True):
    labels = {f"{vendor}/gpu.product": product} if labeled else {}
    status = NodeStatus()
    if allocatable:
        status.allocatable = {f"{vendor}/gpu": str(gpus)}
    else:
        status.capacity = {f"{vendor}/gpu": str(gpus)}
    return client.create(
        Node(metadata=ObjectMeta(name=name, namespace="", labels=labels), status=status)
    )


class TestCollectInventory:
    def test_aggregates_across_nodes(self):
        client = InMemoryKubeClient()
        make_node(client, "n0", gpus=8)
        make_node(client, "n1", gpus=8)
        make_node(client, "n2", product="MI300X", gpus=4)
        inv = collector.collect_inventory_k8s(client)
        assert inv["MI355X"] == {"count": 16, "nodes": 2, "vendor": "amd.com"}
        assert inv["MI300X"]["count"] == 4

    def test_capacity_fallback_when_allocatable_missing(self):
        client = InMemoryKubeClient()
        make_node(client, "n0", gpus=8, allocatable=False)
        inv = collector.collect_inventory_k8s(client)
        assert inv["MI355X"]["count"] == 8

    def test_nodes_without_gpus_or_labels_skipped(self):
        client = InMemoryKubeClient()
        client.create(Node(metadata=ObjectMeta(name="cpu-node", namespace="")))
        make_node(client, "unlabeled", gpus=8, labeled=False)
        inv = collector.collect_inventory_k8s(client)
        assert inv == {}

    def test_multi_vendor_amd_first(self):
        client = InMemoryKubeClient()
        make_node(client, "amd0", gpus=8)
        make_node(client, "nv0", product="H100", gpus=8, vendor="nvidia.com")
        inv = collector.collect_inventory_k8s(client)
        assert inv["MI355X"]["vendor"] == "amd.com"
        assert inv["H100"]["vendor"] == "nvidia.com"

    def test_multiple_gpu_types_on_same_node(self):
        # collector_test.go:182 — one node exposing two vendors' extended
        # resources yields two inventory entries, each with nodes=1
        client = InMemoryKubeClient()
        n = Node(
            metadata=ObjectMeta(
                name="mixed", namespace="",
                labels={
                    "amd.com/gpu.product": "MI355X",
                    "nvidia.com/gpu.product": "H100",
                },
            ),
            status=NodeStatus(
                allocatable={"amd.com/gpu": "8", "nvidia.com/gpu": "4"}
            ),
        )
        client.create(n)
        inv = collector.collect_inventory_k8s(client)
        assert inv["MI355X"] == {"count": 8, "nodes": 1, "vendor": "amd.com"}
        assert inv["H100"] == {"count": 4, "nodes": 1, "vendor": "nvidia.com"}

    def test_unparseable_count_skipped(self):
        client = InMemoryKubeClient()
        n = Node(
            metadata=ObjectMeta(name="bad", namespace="",
                                labels={"amd.com/gpu.product": "MI355X"}),
            status=NodeStatus(allocatable={"amd.com/gpu": "eight"}),
        )
        client.create(n)
        assert collector.collect_inventory_k8s(client) == {}


class TestInventoryIntoCapacity:
    def test_inventory_overrides_static_capacity(self):
        acc_cm = {
            "MI355X": {"device": "AMD-MI355X-288GB", "cost": "85.0", "capacity": "99"},
            "MI300X": {"device": "AMD-MI300X-192GB", "cost": "65.0"},
        }
        inv = {"MI355X": {"count": 16, "nodes": 2, "vendor": "amd.com"}}
        sd = create_system_data(acc_cm, {}, {"WVA_OPTIMIZER_MODE": "limited"},
                                inventory=inv)
        counts = {c.type: c.count for c in sd.spec.capacity.count}
        # live inventory (16) wins over the static field (99)
        assert counts["AMD-MI355X-288GB"] == 16
        assert "AMD-MI300X-192GB" not in counts  # no static field, no nodes


class TestInventoryThroughReconciler:
    def test_limited_mode_capacity_from_nodes(self):
        """End to end: WVA_INVENTORY=k8s + two 8-GPU MI355X nodes cap a
        demand that would otherwise exceed the pool."""
        cluster = make_cluster()
        cm = cluster.get(ConfigMap, CONFIG_MAP_NAME, CONFIG_MAP_NAMESPACE)
        cm.data["WVA_OPTIMIZER_MODE"] = "limited"
        cm.data["WVA_INVENTORY"] = "k8s"
        cm.data["WVA_SATURATION_POLICY"] = "PriorityExhaustive"
        cm.data["WVA_DELAYED_BEST_EFFORT"] = "true"
        cluster.update(cm)
        # one node with 2 GPUs: demand below will want more replicas
        make_node(cluster, "node-0", gpus=2)

        make_deployment(cluster, replicas=1)
        # binding ITL SLO forces ~3+ replicas of demand at this load
        make_va(cluster, max_batch=16, alpha="12.0", beta="6.0",
                gamma="4.0", delta="0.01")
        prom = MockPromAPI()
        set_load_metrics(prom, "default/llama-8b", "default",
                         arrival_rps=8.0, in_tokens=32.0, out_tokens=25.0)

        registry = CollectorRegistry()
        ctrl_metrics.init_metrics(registry)
        try:
            VariantAutoscalingReconciler(cluster, prom).reconcile()
        finally:
            ctrl_metrics.reset_metrics()
        va = cluster.get(v1alpha1.VariantAutoscaling, "vllm-llama", "default")
        desired = va.status.desired_optimized_alloc.num_replicas
        # the 2-GPU node pool caps the allocation at 2 replicas (accCount 1)
        assert 1 <= desired <= 2, f"node pool must cap replicas, got {desired}"

    def test_unlimited_mode_ignores_inventory(self):
        cluster = make_cluster()
        make_node(cluster, "node-0", gpus=1)
        make_deployment(cluster, replicas=1)
        make_va(cluster, max_batch=16, alpha="12.0", beta="6.0",
                gamma="4.0", delta="0.01")
        prom = MockPromAPI()
        set_load_metrics(prom, "default/llama-8b", "default",
                         arrival_rps=8.0, in_tokens=32.0, out_tokens=25.0)
        registry = CollectorRegistry()
        ctrl_metrics.init_metrics(registry)
        try:
            VariantAutoscalingReconciler(cluster, prom).reconcile()
        finally:
            ctrl_metrics.reset_metrics()
        va = cluster.get(v1alpha1.VariantAutoscaling, "vllm-llama", "default")
        # default unlimited mode: the 1-GPU node must not constrain
        assert va.status.desired_optimized_alloc.num_replicas >= 3


class TestNodeOverHTTP:
    def test_node_crud_and_inventory_via_http(self):
        import threading
        import time as _time

        import uvicorn

        from wva_amd.kube.http_client import HTTPKubeClient
        from wva_amd.kube.stub_server import create_stub_api_server

        store = InMemoryKubeClient()
        app, _ = create_stub_api_server(store)
        server = uvicorn.Server(
            uvicorn.Config(app, host="127.0.0.1", port=0, log_level="error")
        )
        t = threading.Thread(target=server.run, daemon=True)
        t.start()
        for _ in range(200):
            if server.started:
                break
            _time.sleep(0.05)
        port = server.servers[0].sockets[0].getsockname()[1]
        client = HTTPKubeClient(base_url=f"http://127.0.0.1:{port}")
        try:
            client.create(
                Node(
                    metadata=ObjectMeta(name="hn0", namespace="",
                                        labels={"amd.com/gpu.product": "MI355X"}),
                    status=NodeStatus(allocatable={"amd.com/gpu": "8"}),
                )
            )
            got = client.get(Node, "hn0", "")
            assert got.status.allocatable["amd.com/gpu"] == "8"
            inv = collector.collect_inventory_k8s(client)
            assert inv["MI355X"]["count"] == 8
            client.delete(Node, "hn0", "")
        finally:
            server.should_exit = True
            t.join(timeout=5)
