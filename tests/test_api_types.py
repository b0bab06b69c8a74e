"""API type tests (mirrors api/v1alpha1/variantautoscaling_types_test.go:
validation behavior, condition transition semantics, serialization)."""

import pytest
from pydantic import ValidationError

from wva_amd.api import v1alpha1
from wva_amd.api.v1alpha1.types import ObjectMeta


def make_va(**kw):
    return v1alpha1.VariantAutoscaling(
        metadata=ObjectMeta(name="va", namespace="ns", generation=3), **kw
    )


class TestValidation:
    def test_numeric_string_pattern_enforced(self):
        with pytest.raises(ValidationError):
            v1alpha1.Allocation(variantCost="not-a-number")
        with pytest.raises(ValidationError):
            v1alpha1.Allocation(itlAverage="-3.0")  # negative not allowed
        a = v1alpha1.Allocation(variantCost="12.50", itlAverage="0", ttftAverage="3.14")
        assert a.variant_cost == "12.50"

    def test_replicas_non_negative(self):
        with pytest.raises(ValidationError):
            v1alpha1.Allocation(numReplicas=-1)
        with pytest.raises(ValidationError):
            v1alpha1.OptimizedAlloc(numReplicas=-2)

    def test_accelerator_profile_bounds(self):
        with pytest.raises(ValidationError):
            v1alpha1.AcceleratorProfile(acc="", accCount=1, maxBatchSize=1)
        with pytest.raises(ValidationError):
            v1alpha1.AcceleratorProfile(acc="MI355X", accCount=0, maxBatchSize=1)
        with pytest.raises(ValidationError):
            v1alpha1.AcceleratorProfile(acc="MI355X", accCount=1, maxBatchSize=0)

    def test_model_profile_needs_accelerator(self):
        with pytest.raises(ValidationError):
            v1alpha1.ModelProfile(accelerators=[])

    def test_camel_case_serialization(self):
        va = make_va()
        d = va.to_dict()
        assert d["apiVersion"] == "llmd.ai/v1alpha1"
        assert d["kind"] == "VariantAutoscaling"
        assert "modelID" in d["spec"]
        assert "currentAlloc" in d["status"]
        assert "desiredOptimizedAlloc" in d["status"]
        # round trip
        va2 = v1alpha1.VariantAutoscaling.model_validate(d)
        assert va2.to_dict() == d


class TestConditions:
    def test_set_and_get(self):
        va = make_va()
        v1alpha1.set_condition(
            va, v1alpha1.TYPE_METRICS_AVAILABLE, "True", v1alpha1.REASON_METRICS_FOUND, "ok"
        )
        c = v1alpha1.get_condition(va, v1alpha1.TYPE_METRICS_AVAILABLE)
        assert c is not None
        assert c.status == "True"
        assert c.observed_generation == 3
        assert c.last_transition_time is not None
        assert v1alpha1.is_condition_true(va, v1alpha1.TYPE_METRICS_AVAILABLE)
        assert not v1alpha1.is_condition_false(va, v1alpha1.TYPE_METRICS_AVAILABLE)

    def test_transition_time_only_changes_on_status_flip(self):
        va = make_va()
        v1alpha1.set_condition(va, "T", "True", "R1", "m1")
        t1 = v1alpha1.get_condition(va, "T").last_transition_time
        # same status, different reason: transition time preserved
        v1alpha1.set_condition(va, "T", "True", "R2", "m2")
        c = v1alpha1.get_condition(va, "T")
        assert c.last_transition_time == t1
        assert c.reason == "R2"
        # status flip: transition time moves forward
        v1alpha1.set_condition(va, "T", "False", "R3", "m3")
        c = v1alpha1.get_condition(va, "T")
        assert c.last_transition_time >= t1
        assert c.status == "False"
        assert v1alpha1.is_condition_false(va, "T")

    def test_multiple_condition_types_coexist(self):
        va = make_va()
        v1alpha1.set_condition(va, v1alpha1.TYPE_METRICS_AVAILABLE, "True", "r", "m")
        v1alpha1.set_condition(va, v1alpha1.TYPE_OPTIMIZATION_READY, "False", "r", "m")
        assert len(va.status.conditions) == 2
        assert v1alpha1.get_condition(va, "missing") is None


class TestDefaults:
    def test_fresh_status(self):
        va = make_va()
        assert va.status.current_alloc.num_replicas == 0
        assert va.status.desired_optimized_alloc.accelerator == ""
        assert va.status.actuation.applied is False
        assert va.status.conditions == []

    def test_identity_constants(self):
        assert v1alpha1.GROUP == "llmd.ai"
        assert v1alpha1.VERSION == "v1alpha1"
        assert v1alpha1.SHORT_NAME == "va"


class TestCRDSchemaDoc:
    def test_generated_doc_in_sync(self):
        """docs/user-guide/crd-schema.md must match the pydantic types
        (regenerate with `python hack/gen_crd_docs.py --write`)."""
        import sys
        from pathlib import Path

        root = Path(__file__).resolve().parent.parent
        sys.path.insert(0, str(root / "hack"))
        import gen_crd_docs

        assert gen_crd_docs.DOC_PATH.read_text() == gen_crd_docs.render()


class TestDeepCopyIndependence:
    """variantautoscaling_types_test.go:97 — mutated copies must never
    alias the original's nested fields (pydantic model_copy(deep=True)
    stands in for the generated DeepCopy)."""

    def _valid_va(self):
        return v1alpha1.VariantAutoscaling.model_validate({
            "metadata": {"name": "a", "namespace": "ns"},
            "spec": {
                "modelID": "model-123",
                "sloClassRef": {"name": "slo-config", "key": "k"},
                "modelProfile": {"accelerators": [{
                    "acc": "MI355X", "accCount": 1, "maxBatchSize": 8,
                    "perfParms": {"decodeParms": {"alpha": "1", "beta": "2"},
                                  "prefillParms": {"gamma": "3", "delta": "4"}},
                }]},
            },
        })

    def test_deep_copy_independence(self):
        orig = self._valid_va()
        cp = orig.model_copy(deep=True)
        cp.spec.model_id = "model-456"
        cp.spec.slo_class_ref.name = "slo-config-2"
        cp.spec.model_profile.accelerators[0].acc = "MI300X"
        cp.status.current_alloc.load.arrival_rate = "20"
        assert orig.spec.model_id == "model-123"
        assert orig.spec.slo_class_ref.name == "slo-config"
        assert orig.spec.model_profile.accelerators[0].acc == "MI355X"
        assert orig.status.current_alloc.load.arrival_rate != "20"

    def test_json_round_trip_exact(self):
        # :120 TestJSONRoundTrip incl. the RFC3339 lastRunTime instant
        import datetime
        import json

        orig = self._valid_va()
        orig.status.desired_optimized_alloc.accelerator = "MI355X"
        orig.status.desired_optimized_alloc.last_run_time = datetime.datetime(
            2026, 9, 14, 12, 0, 0, tzinfo=datetime.timezone.utc
        )
        raw = json.dumps(orig.model_dump(by_alias=True, exclude_none=True, mode="json"))
        back = v1alpha1.VariantAutoscaling.model_validate(json.loads(raw))
        assert (
            back.status.desired_optimized_alloc.last_run_time
            == orig.status.desired_optimized_alloc.last_run_time
        )
        assert back.model_dump(by_alias=True, exclude_none=True, mode="json") == \
            orig.model_dump(by_alias=True, exclude_none=True, mode="json")

    def test_last_run_time_serializes_rfc3339(self):
        # :228 TestOptimizedAllocLastRunTimeJSON
        import datetime

        va = self._valid_va()
        va.status.desired_optimized_alloc.last_run_time = datetime.datetime(
            2026, 9, 14, 12, 0, 0, tzinfo=datetime.timezone.utc
        )
        d = va.model_dump(by_alias=True, exclude_none=True, mode="json")
        raw = d["status"]["desiredOptimizedAlloc"]["lastRunTime"]
        assert isinstance(raw, str) and raw.startswith("2026-09-14T12:00:00")

    def test_status_present_with_zero_defaults(self):
        # :166 TestStatusOmitEmpty — a freshly-specced VA still serializes a
        # status object whose leaves carry zero values; flipping one bool
        # keeps the key present
        va = self._valid_va()
        d = va.model_dump(by_alias=True, exclude_none=True, mode="json")
        assert "status" in d
        assert d["status"]["currentAlloc"]["accelerator"] in ("", "MI355X")
        fresh = v1alpha1.VariantAutoscaling(
            metadata=va.metadata, spec=va.spec
        ).model_dump(by_alias=True, exclude_none=True, mode="json")
        assert fresh["status"]["desiredOptimizedAlloc"]["numReplicas"] == 0
        assert fresh["status"]["actuation"]["applied"] is False
        va2 = v1alpha1.VariantAutoscaling(metadata=va.metadata, spec=va.spec)
        va2.status.actuation.applied = True
        d2 = va2.model_dump(by_alias=True, exclude_none=True, mode="json")
        assert d2["status"]["actuation"]["applied"] is True
