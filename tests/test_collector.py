"""Collector suite (mirrors internal/collector tests: query/status tests
with MockPromAPI, unit conversions, status string formats)."""


from wva_amd.api import v1alpha1
from wva_amd.api.v1alpha1.types import ObjectMeta
from wva_amd.controller import collector
from wva_amd.controller.promclient import MockPromAPI
from wva_amd.kube import Deployment, DeploymentSpec, DeploymentStatus
from kube_fixtures import make_va, set_load_metrics
from wva_amd.kube import InMemoryKubeClient


def make_deploy(replicas=3, namespace="prod"):
    return Deployment(
        metadata=ObjectMeta(name="d", namespace=namespace),
        spec=DeploymentSpec(replicas=replicas),
        status=DeploymentStatus(replicas=replicas),
    )


def make_test_va(model="m/llama", acc="MI355X"):
    client = InMemoryKubeClient()
    return make_va(client, name="d", namespace="prod", model_id=model, accelerator=acc)


class TestAddMetricsToOptStatus:
    def test_unit_conversions_and_formats(self):
        prom = MockPromAPI()
        set_load_metrics(
            prom, "m/llama", "prod",
            arrival_rps=2.5, in_tokens=100.25, out_tokens=50.5, ttft_s=0.125, itl_s=0.0301,
        )
        alloc = collector.add_metrics_to_opt_status(
            make_test_va(), make_deploy(replicas=3), accelerator_cost=85.0, prom=prom
        )
        # req/s -> req/min
        assert alloc.load.arrival_rate == "150.00"
        # s -> ms with 2-decimal strings (CRD numeric-string pattern)
        assert alloc.ttft_average == "125.00"
        assert alloc.itl_average == "30.10"
        assert alloc.load.avg_input_tokens == "100.25"
        assert alloc.load.avg_output_tokens == "50.50"
        # replicas x unit cost
        assert alloc.variant_cost == "255.00"
        assert alloc.num_replicas == 3
        assert alloc.max_batch == 256  # hardcoded parity (collector.go:258)
        assert alloc.accelerator == "MI355X"
        # the allocation validates against the CRD patterns
        v1alpha1.Allocation.model_validate(alloc.to_dict())

    def test_missing_accelerator_label_warns_but_proceeds(self):
        prom = MockPromAPI()
        va = make_test_va()
        va.metadata.labels = {}
        alloc = collector.add_metrics_to_opt_status(va, make_deploy(), 85.0, prom)
        assert alloc.accelerator == ""

    def test_nan_metrics_become_zero(self):
        prom = MockPromAPI()
        prom.set_result(collector.arrival_query("m/llama", "prod"), float("nan"))
        alloc = collector.add_metrics_to_opt_status(make_test_va(), make_deploy(), 85.0, prom)
        assert alloc.load.arrival_rate == "0.00"


class TestRateWindowEnv:
    def test_window_override(self, monkeypatch):
        monkeypatch.setenv("WVA_RATE_WINDOW", "5m")
        assert "[5m]" in collector.arrival_query("m", "ns")
        assert "[5m]" in collector.itl_query("m", "ns")
        monkeypatch.delenv("WVA_RATE_WINDOW")
        assert "[1m]" in collector.arrival_query("m", "ns")


class TestInventoryStub:
    def test_stub_and_vendor_order(self):
        assert collector.collect_inventory_k8s(None) == {}
        assert collector.VENDORS[0] == "amd.com"  # AMD first-class
