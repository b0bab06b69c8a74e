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
