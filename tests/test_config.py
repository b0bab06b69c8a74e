"""Spec round-trip and enum tests (mirrors pkg/config/config_test.go)."""

import json

from wva_amd.config import (
    SaturationPolicy,
    SystemSpec,
    mi355x_accelerator_configmap,
)
from fixtures import make_spec


class TestSaturationPolicy:
    def test_round_trip(self):
        for p in SaturationPolicy:
            assert SaturationPolicy.parse(str(p)) is p

    def test_unknown_defaults_to_none(self):
        assert SaturationPolicy.parse("bogus") is SaturationPolicy.NONE
        assert SaturationPolicy.parse("") is SaturationPolicy.NONE


class TestSpecRoundTrip:
    def test_json_round_trip(self):
        spec = make_spec(capacity=[("AMD-MI355X-288GB", 16)])
        text = spec.to_json()
        spec2 = SystemSpec.from_json(text)
        assert spec2.to_dict() == spec.to_dict()

    def test_json_field_names(self):
        spec = make_spec()
        d = spec.to_dict()
        # reference-compatible key names (pkg/config/types.go)
        assert "acceleratorData" in d and "accelerators" in d["acceleratorData"]
        assert "modelData" in d and "models" in d["modelData"]
        assert "serviceClassData" in d
        acc = d["acceleratorData"]["accelerators"][0]
        assert {"name", "type", "multiplicity", "memSize", "memBW", "power", "cost"} <= set(acc)
        mt = d["serviceClassData"]["serviceClasses"][0]["modelTargets"][0]
        assert {"model", "slo-itl", "slo-ttft", "slo-tps"} <= set(mt)
        srv = d["serverData"]["servers"][0]
        assert "class" in srv and "currentAlloc" in srv
        assert srv["currentAlloc"]["load"]["arrivalRate"] == 60.0

    def test_defaults_from_empty(self):
        spec = SystemSpec.from_dict({})
        assert spec.accelerators.spec == []
        assert spec.optimizer.spec.unlimited is False


class TestMI355XCatalog:
    def test_configmap_shape(self):
        cm = mi355x_accelerator_configmap()
        assert "MI355X" in cm
        entry = json.loads(cm["MI355X"])
        assert entry["device"] == "AMD-MI355X-288GB"
        assert float(entry["cost"]) > 0
        assert entry["memSize"] == "288"


class TestPackageAlias:
    def test_full_name_alias(self):
        import workload_variant_autoscaler_amd as full
        import workload_variant_autoscaler_amd.analyzer as full_analyzer
        import wva_amd
        import wva_amd.analyzer

        assert full is wva_amd
        assert full_analyzer is wva_amd.analyzer
