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


class TestValidateMetricsAvailability:
    """collector_test.go:410-530 validation scenarios, as a table."""

    MODEL, NS = "m/llama", "prod"

    def _q(self, with_ns=True):
        from wva_amd.controller import constants

        if with_ns:
            return (
                f'{constants.VLLM_REQUEST_SUCCESS_TOTAL}'
                f'{{{constants.LABEL_MODEL_NAME}="{self.MODEL}",'
                f'{constants.LABEL_NAMESPACE}="{self.NS}"}}'
            )
        return (
            f'{constants.VLLM_REQUEST_SUCCESS_TOTAL}'
            f'{{{constants.LABEL_MODEL_NAME}="{self.MODEL}"}}'
        )

    def test_available_with_namespace_label(self):
        prom = MockPromAPI()
        prom.set_result(self._q(), 5.0)
        r = collector.validate_metrics_availability(prom, self.MODEL, self.NS)
        assert r.available

    def test_fallback_without_namespace_label(self):
        # emulator scrape: first query empty, fallback (no namespace)
        # has the series -> available
        prom = MockPromAPI()
        prom.query_results[self._q()] = []
        prom.set_result(self._q(with_ns=False), 5.0)
        r = collector.validate_metrics_availability(prom, self.MODEL, self.NS)
        assert r.available

    def test_unavailable_on_prometheus_error(self):
        prom = MockPromAPI()
        prom.set_error(self._q(), RuntimeError("boom"))
        r = collector.validate_metrics_availability(prom, self.MODEL, self.NS)
        assert not r.available
        assert r.reason == v1alpha1.REASON_PROMETHEUS_ERROR

    def test_unavailable_when_no_metrics_found(self):
        prom = MockPromAPI()
        prom.query_results[self._q()] = []
        prom.query_results[self._q(with_ns=False)] = []
        r = collector.validate_metrics_availability(prom, self.MODEL, self.NS)
        assert not r.available
        assert r.reason == v1alpha1.REASON_METRICS_MISSING
        # the message carries the operator checklist
        assert "ServiceMonitor" in r.message

    def test_unavailable_when_metrics_stale(self):
        prom = MockPromAPI()
        prom.set_result(self._q(), 5.0, age_seconds=301.0)  # > 5 min
        r = collector.validate_metrics_availability(prom, self.MODEL, self.NS)
        assert not r.available
        assert r.reason == v1alpha1.REASON_METRICS_STALE

    def test_fallback_query_error_is_prometheus_error(self):
        prom = MockPromAPI()
        prom.query_results[self._q()] = []
        prom.set_error(self._q(with_ns=False), RuntimeError("fallback boom"))
        r = collector.validate_metrics_availability(prom, self.MODEL, self.NS)
        assert not r.available
        assert r.reason == v1alpha1.REASON_PROMETHEUS_ERROR

    def test_fresh_metrics_within_window_accepted(self):
        prom = MockPromAPI()
        prom.set_result(self._q(), 5.0, age_seconds=299.0)  # just inside
        r = collector.validate_metrics_availability(prom, self.MODEL, self.NS)
        assert r.available


class TestFixValueTable:
    """collector_test.go:372-408 FixValue table."""

    def test_table(self):
        import math

        fix = collector._fix_value
        assert fix(float("nan")) == 0.0
        assert fix(float("inf")) == 0.0
        assert fix(float("-inf")) == 0.0
        assert fix(42.5) == 42.5
        assert fix(0.0) == 0.0
        assert fix(-3.25) == -3.25  # negatives pass through unchanged
