"""Structural-schema validation for CustomResources.

Implements the subset of OpenAPI v3 that Kubernetes structural schemas
use (and that the VariantAutoscaling CRD exercises): type, required,
properties, items, additionalProperties, enum, pattern, minLength,
maxLength, minimum, minItems, minProperties.  The stub API server
validates VA create/update/status writes through this against the
*shipped CRD YAML* (deploy/crd/llmd.ai_variantautoscalings.yaml), so a
spec the real apiserver would reject is rejected here too — the envtest
analog of admission-time schema validation
(/root/reference/internal/controller/suite_test.go:56-93 installs the
generated CRD for the same reason).
"""

from __future__ import annotations

import re
from pathlib import Path
from typing import Any, Dict, List, Optional

import yaml

CRD_PATH = (
    Path(__file__).resolve().parent.parent.parent
    / "deploy"
    / "crd"
    / "llmd.ai_variantautoscalings.yaml"
)


class SchemaValidationError(Exception):
    """Aggregate of field-path-qualified schema violations (422 payload)."""

    def __init__(self, causes: List[str]) -> None:
        self.causes = causes
        super().__init__("; ".join(causes))


def load_crd_schema(path: Path = CRD_PATH) -> Dict[str, Any]:
    """The openAPIV3Schema of the served version from the CRD manifest."""
    crd = yaml.safe_load(path.read_text())
    for version in crd["spec"]["versions"]:
        if version.get("served"):
            return version["schema"]["openAPIV3Schema"]
    raise ValueError(f"no served version in {path}")


def _type_ok(value: Any, typ: str) -> bool:
    if typ == "object":
        return isinstance(value, dict)
    if typ == "array":
        return isinstance(value, list)
    if typ == "string":
        return isinstance(value, str)
    if typ == "integer":
        return isinstance(value, int) and not isinstance(value, bool)
    if typ == "number":
        return isinstance(value, (int, float)) and not isinstance(value, bool)
    if typ == "boolean":
        return isinstance(value, bool)
    return True


def _validate(value: Any, schema: Dict[str, Any], path: str, errs: List[str]) -> None:
    typ = schema.get("type")
    if value is None:
        # serializers emit explicit nulls for unset optionals; the real
        # server prunes them rather than failing type validation
        return
    if typ and not _type_ok(value, typ):
        errs.append(f"{path}: expected {typ}, got {type(value).__name__}")
        return

    if "enum" in schema and value not in schema["enum"]:
        errs.append(f"{path}: {value!r} not in {schema['enum']}")

    if typ == "string":
        if "minLength" in schema and len(value) < schema["minLength"]:
            errs.append(f"{path}: shorter than minLength {schema['minLength']}")
        if "maxLength" in schema and len(value) > schema["maxLength"]:
            errs.append(f"{path}: longer than maxLength {schema['maxLength']}")
        if "pattern" in schema and not re.search(schema["pattern"], value):
            errs.append(f"{path}: {value!r} does not match {schema['pattern']!r}")

    if typ in ("integer", "number"):
        if "minimum" in schema and value < schema["minimum"]:
            errs.append(f"{path}: {value} below minimum {schema['minimum']}")
        if "maximum" in schema and value > schema["maximum"]:
            errs.append(f"{path}: {value} above maximum {schema['maximum']}")

    if typ == "object":
        for req in schema.get("required", []):
            if req not in value:
                errs.append(f"{path}.{req}: required value missing")
        if "minProperties" in schema and len(value) < schema["minProperties"]:
            errs.append(
                f"{path}: fewer than minProperties {schema['minProperties']}"
            )
        props = schema.get("properties", {})
        addl = schema.get("additionalProperties")
        for key, sub in value.items():
            if key in props:
                _validate(sub, props[key], f"{path}.{key}", errs)
            elif isinstance(addl, dict):
                _validate(sub, addl, f"{path}.{key}", errs)
            # unknown fields are pruned, not rejected, by structural
            # schemas without x-kubernetes-preserve-unknown-fields

    if typ == "array":
        if "minItems" in schema and len(value) < schema["minItems"]:
            errs.append(f"{path}: fewer than minItems {schema['minItems']}")
        items = schema.get("items")
        if isinstance(items, dict):
            for i, sub in enumerate(value):
                _validate(sub, items, f"{path}[{i}]", errs)


class CRDValidator:
    """Validates object dicts against the CRD's served-version schema."""

    def __init__(self, schema: Optional[Dict[str, Any]] = None) -> None:
        self.schema = schema if schema is not None else load_crd_schema()

    def validate(self, obj: Dict[str, Any], *, subresource: str = "") -> None:
        """Raise SchemaValidationError on violations.

        ``subresource="status"`` validates the whole object (the real
        status endpoint does — only *changes* are restricted to status).
        """
        errs: List[str] = []
        _validate(obj, self.schema, "", errs)
        if errs:
            raise SchemaValidationError(errs)
