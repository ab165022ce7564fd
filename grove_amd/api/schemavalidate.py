"""Structural-schema admission: server-side defaulting + type validation for the CRD
kinds, driven by the SAME schemas the CRD YAML is rendered from (api/openapi.py).

Parity role: on a real cluster the kube-apiserver enforces the CRD's structural
openAPIV3Schema (types, enums, bounds, required, defaults) before any webhook runs;
the round-1 store accepted any shape ("no server-side type/enum/
bounds validation exists at the API layer"). This module walks an object against the
schema: `apply_defaults` fills schema `default:` values (structural defaulting) and
`validate` rejects type violations with kube-style field paths.

CEL x-kubernetes-validations rules are not executed here; their semantic content
(topology pack requiredness, packDomain deprecation) is enforced by
api/validation.py's TopologyConstraintValidator.

Scope: the `spec` subtree (user input). Status subresources are written by
controllers and validated by their own writers.
"""
from __future__ import annotations

import re
from typing import Any, Dict, List, Optional

from ..kubecore.store import invalid
from .openapi import schemas

Schema = Dict[str, Any]

_INT32_MIN, _INT32_MAX = -2**31, 2**31 - 1
_INT64_MIN, _INT64_MAX = -2**63, 2**63 - 1


def apply_defaults(obj: Any, schema: Schema) -> None:
    """Fill `default:` values for absent properties, recursively (structural
    defaulting, same semantics as the apiserver's CRD defaulting)."""
    if not isinstance(obj, dict):
        return
    props = schema.get("properties") or {}
    for name, sub in props.items():
        if name not in obj and "default" in sub:
            obj[name] = _copy_default(sub["default"])
        if name in obj:
            val = obj[name]
            if isinstance(val, dict):
                apply_defaults(val, sub)
            elif isinstance(val, list) and "items" in sub:
                for item in val:
                    apply_defaults(item, sub["items"])


def _copy_default(v: Any) -> Any:
    if isinstance(v, (dict, list)):
        import copy
        return copy.deepcopy(v)
    return v


def validate(obj: Any, schema: Schema, path: str = "") -> List[str]:
    errors: List[str] = []
    _walk(obj, schema, path, errors)
    return errors


def _walk(val: Any, schema: Schema, path: str, errors: List[str]) -> None:
    if schema.get("x-kubernetes-int-or-string"):
        if not isinstance(val, (str, int)) or isinstance(val, bool):
            errors.append(f"{path}: expected int-or-string, got "
                          f"{type(val).__name__}")
        return
    if schema.get("x-kubernetes-preserve-unknown-fields") and "type" not in schema:
        return
    t = schema.get("type")
    if t == "object":
        if not isinstance(val, dict):
            errors.append(f"{path}: expected object, got {type(val).__name__}")
            return
        for req in schema.get("required") or []:
            if req not in val:
                errors.append(f"{path}.{req}: required field is missing")
        props = schema.get("properties") or {}
        addl = schema.get("additionalProperties")
        for k, v in val.items():
            sub = props.get(k)
            if sub is not None:
                _walk(v, sub, f"{path}.{k}", errors)
            elif isinstance(addl, dict):
                _walk(v, addl, f"{path}.{k}", errors)
            # unknown fields: tolerated (the apiserver prunes; controllers may
            # carry internal annotations) — type safety covers declared fields
    elif t == "array":
        if not isinstance(val, list):
            errors.append(f"{path}: expected array, got {type(val).__name__}")
            return
        min_items = schema.get("minItems")
        if min_items is not None and len(val) < min_items:
            errors.append(f"{path}: must contain at least {min_items} items")
        items = schema.get("items")
        if isinstance(items, dict):
            for i, item in enumerate(val):
                _walk(item, items, f"{path}[{i}]", errors)
    elif t == "string":
        if not isinstance(val, str):
            errors.append(f"{path}: expected string, got {type(val).__name__}")
            return
        _check_string(val, schema, path, errors)
    elif t == "integer":
        if isinstance(val, bool) or not isinstance(val, int):
            errors.append(f"{path}: expected integer, got {type(val).__name__}")
            return
        lo, hi = ((_INT32_MIN, _INT32_MAX) if schema.get("format") == "int32"
                  else (_INT64_MIN, _INT64_MAX))
        if not (lo <= val <= hi):
            errors.append(f"{path}: {val} out of {schema.get('format', 'int')} range")
        _check_bounds(val, schema, path, errors)
    elif t == "number":
        if isinstance(val, bool) or not isinstance(val, (int, float)):
            errors.append(f"{path}: expected number, got {type(val).__name__}")
            return
        _check_bounds(val, schema, path, errors)
    elif t == "boolean":
        if not isinstance(val, bool):
            errors.append(f"{path}: expected boolean, got {type(val).__name__}")


def _check_string(val: str, schema: Schema, path: str, errors: List[str]) -> None:
    enum = schema.get("enum")
    if enum and val not in enum:
        errors.append(f"{path}: unsupported value {val!r}; supported values: "
                      + ", ".join(repr(e) for e in enum))
    max_len = schema.get("maxLength")
    if max_len is not None and len(val) > max_len:
        errors.append(f"{path}: may not be longer than {max_len} characters")
    min_len = schema.get("minLength")
    if min_len is not None and len(val) < min_len:
        errors.append(f"{path}: may not be shorter than {min_len} characters")
    pattern = schema.get("pattern")
    if pattern and not re.match(pattern, val):
        errors.append(f"{path}: {val!r} does not match pattern {pattern!r}")
    if schema.get("format") == "date-time" and val:
        if not re.match(r"^\d{4}-\d{2}-\d{2}T\d{2}:\d{2}:\d{2}", val):
            errors.append(f"{path}: {val!r} is not RFC3339 date-time")


def _check_bounds(val, schema: Schema, path: str, errors: List[str]) -> None:
    minimum = schema.get("minimum")
    if minimum is not None and val < minimum:
        errors.append(f"{path}: must be greater than or equal to {minimum}")
    maximum = schema.get("maximum")
    if maximum is not None and val > maximum:
        errors.append(f"{path}: must be less than or equal to {maximum}")


class StructuralSchemaAdmission:
    """Per-kind mutator (defaults) + validator (types) registered on the store."""

    def __init__(self) -> None:
        self._schemas = schemas()

    def register(self, store) -> None:
        for kind, schema in self._schemas.items():
            spec_schema = schema["properties"]["spec"]
            store.register_mutator(kind, self._defaulter(spec_schema))
            store.register_validator(kind, self._validator(kind, spec_schema))

    def _defaulter(self, spec_schema: Schema):
        def default_fn(obj: Dict[str, Any],
                       _old: Optional[Dict[str, Any]] = None) -> None:
            if isinstance(obj.get("spec"), dict):
                apply_defaults(obj["spec"], spec_schema)
        return default_fn

    def _validator(self, kind: str, spec_schema: Schema):
        def validate_fn(obj: Dict[str, Any],
                        old: Optional[Dict[str, Any]]) -> None:
            if "spec" not in obj:
                raise invalid(f"{kind}: spec is required")
            errors = validate(obj["spec"], spec_schema, "spec")
            if errors:
                raise invalid(
                    f"{kind} schema violation: " + "; ".join(errors[:10]))
        return validate_fn
