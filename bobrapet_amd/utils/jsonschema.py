"""Minimal JSON Schema validation + defaults application.

Covers the subset the reference exercises through gojsonschema
(reference: internal/controller/runs/schema_validation.go,
pkg/runs/inputs/defaults.go:15): type, required, properties,
additionalProperties, enum, items, min/max, minLength/maxLength,
minItems/maxItems, pattern, and top-down ``default`` injection.
"""
from __future__ import annotations

import re
import typing as _t

_TYPE_CHECKS = {
    "object": lambda v: isinstance(v, dict),
    "array": lambda v: isinstance(v, list),
    "string": lambda v: isinstance(v, str),
    "integer": lambda v: isinstance(v, int) and not isinstance(v, bool),
    "number": lambda v: isinstance(v, (int, float)) and not isinstance(v, bool),
    "boolean": lambda v: isinstance(v, bool),
    "null": lambda v: v is None,
}


def validate_instance(value, schema: dict, path: str = "$") -> _t.List[str]:
    """Validate ``value`` against ``schema``; returns a list of error strings."""
    errors: _t.List[str] = []
    if not isinstance(schema, dict):
        return errors

    typ = schema.get("type")
    if typ is not None:
        types = typ if isinstance(typ, list) else [typ]
        if not any(_TYPE_CHECKS.get(t, lambda v: True)(value) for t in types):
            errors.append(f"{path}: expected type {typ}, got {type(value).__name__}")
            return errors

    if "enum" in schema and value not in schema["enum"]:
        errors.append(f"{path}: value {value!r} not in enum {schema['enum']}")

    if "const" in schema and value != schema["const"]:
        errors.append(f"{path}: value {value!r} != const {schema['const']!r}")

    if isinstance(value, dict):
        for req in schema.get("required", []):
            if req not in value:
                errors.append(f"{path}: missing required property {req!r}")
        props = schema.get("properties", {})
        for k, v in value.items():
            if k in props:
                errors.extend(validate_instance(v, props[k], f"{path}.{k}"))
            elif schema.get("additionalProperties") is False:
                errors.append(f"{path}: unexpected property {k!r}")
            elif isinstance(schema.get("additionalProperties"), dict):
                errors.extend(
                    validate_instance(v, schema["additionalProperties"], f"{path}.{k}")
                )
        if "minProperties" in schema and len(value) < schema["minProperties"]:
            errors.append(f"{path}: fewer than {schema['minProperties']} properties")
        if "maxProperties" in schema and len(value) > schema["maxProperties"]:
            errors.append(f"{path}: more than {schema['maxProperties']} properties")

    if isinstance(value, list):
        items = schema.get("items")
        if isinstance(items, dict):
            for i, v in enumerate(value):
                errors.extend(validate_instance(v, items, f"{path}[{i}]"))
        if "minItems" in schema and len(value) < schema["minItems"]:
            errors.append(f"{path}: fewer than {schema['minItems']} items")
        if "maxItems" in schema and len(value) > schema["maxItems"]:
            errors.append(f"{path}: more than {schema['maxItems']} items")

    if isinstance(value, str):
        if "minLength" in schema and len(value) < schema["minLength"]:
            errors.append(f"{path}: shorter than minLength {schema['minLength']}")
        if "maxLength" in schema and len(value) > schema["maxLength"]:
            errors.append(f"{path}: longer than maxLength {schema['maxLength']}")
        if "pattern" in schema and not re.search(schema["pattern"], value):
            errors.append(f"{path}: does not match pattern {schema['pattern']!r}")

    if isinstance(value, (int, float)) and not isinstance(value, bool):
        if "minimum" in schema and value < schema["minimum"]:
            errors.append(f"{path}: {value} < minimum {schema['minimum']}")
        if "maximum" in schema and value > schema["maximum"]:
            errors.append(f"{path}: {value} > maximum {schema['maximum']}")
        if "exclusiveMinimum" in schema and value <= schema["exclusiveMinimum"]:
            errors.append(f"{path}: {value} <= exclusiveMinimum {schema['exclusiveMinimum']}")
        if "exclusiveMaximum" in schema and value >= schema["exclusiveMaximum"]:
            errors.append(f"{path}: {value} >= exclusiveMaximum {schema['exclusiveMaximum']}")

    return errors


def apply_defaults(value, schema: dict):
    """Return a copy of ``value`` with schema ``default``s injected for missing
    object properties (reference: pkg/runs/inputs/defaults.go)."""
    if not isinstance(schema, dict):
        return value
    if isinstance(value, dict):
        out = dict(value)
        for k, sub in schema.get("properties", {}).items():
            if k not in out and isinstance(sub, dict) and "default" in sub:
                out[k] = sub["default"]
            elif k in out:
                out[k] = apply_defaults(out[k], sub)
        return out
    if isinstance(value, list) and isinstance(schema.get("items"), dict):
        return [apply_defaults(v, schema["items"]) for v in value]
    return value
