"""OpenAPI v3 structural-schema validator (CRD subset).

Validates decoded JSON objects against the openAPIV3Schema the CRD
generator emits (wva_amd/api/crd.py), the way kube-apiserver validates
custom resources on create/update/patch: the *result* of applying a
patch is validated, which is exactly the interaction behind reference
issue #731 (a JSON merge patch carrying a partial
``status.desiredOptimizedAlloc`` produces a merged object that fails the
nested ``required: [accelerator, numReplicas]`` when the stored object
had no prior alloc — reference workaround at
variantautoscaling_controller.go:237-252).

Supported keywords: type, properties, required, items, enum, pattern,
minLength, maxLength, minimum, maximum, format (date-time checked
loosely), x-kubernetes-* ignored. Unknown fields are allowed (structural
schemas prune rather than reject; we do neither — the serde layer drops
unknown fields already).
"""
from __future__ import annotations

import re
from typing import Any, Dict, List

_DATETIME_RE = re.compile(
    r"^\d{4}-\d{2}-\d{2}T\d{2}:\d{2}:\d{2}(\.\d+)?(Z|[+-]\d{2}:\d{2})$"
)


def validate(schema: Dict[str, Any], obj: Any, path: str = "") -> List[str]:
    """Return a list of validation error strings ([] = valid)."""
    errors: List[str] = []
    _validate(schema, obj, path or "<root>", errors)
    return errors


def _validate(schema: Dict[str, Any], obj: Any, path: str, errors: List[str]) -> None:
    stype = schema.get("type")

    if stype == "object" or "properties" in schema or "required" in schema:
        if obj is None:
            # omitted optional object — presence is the parent's concern
            return
        if not isinstance(obj, dict):
            errors.append(f"{path}: expected object, got {type(obj).__name__}")
            return
        for req in schema.get("required", []):
            if req not in obj or obj[req] is None:
                errors.append(f"{path}.{req}: Required value")
        props = schema.get("properties", {})
        for key, sub in props.items():
            if key in obj and obj[key] is not None:
                _validate(sub, obj[key], f"{path}.{key}", errors)
        addl = schema.get("additionalProperties")
        if isinstance(addl, dict):
            for key, val in obj.items():
                if key not in props and val is not None:
                    _validate(addl, val, f"{path}.{key}", errors)
        return

    if stype == "array":
        if obj is None:
            return
        if not isinstance(obj, list):
            errors.append(f"{path}: expected array, got {type(obj).__name__}")
            return
        item_schema = schema.get("items")
        if isinstance(item_schema, dict):
            for i, item in enumerate(obj):
                _validate(item_schema, item, f"{path}[{i}]", errors)
        return

    if stype == "string":
        if not isinstance(obj, str):
            errors.append(f"{path}: expected string, got {type(obj).__name__}")
            return
        if "minLength" in schema and len(obj) < schema["minLength"]:
            errors.append(
                f"{path}: Invalid value: \"{obj}\": must be at least "
                f"{schema['minLength']} chars long"
            )
        if "maxLength" in schema and len(obj) > schema["maxLength"]:
            errors.append(f"{path}: too long (max {schema['maxLength']})")
        # JSON Schema `pattern` is a PARTIAL match (anchor explicitly
        # with ^...$ for a full match — both CRD patterns here do)
        if "pattern" in schema and not re.search(schema["pattern"], obj):
            errors.append(
                f"{path}: Invalid value: \"{obj}\": must match pattern "
                f"{schema['pattern']}"
            )
        if "enum" in schema and obj not in schema["enum"]:
            errors.append(f"{path}: unsupported value \"{obj}\"")
        if schema.get("format") == "date-time" and obj and not _DATETIME_RE.match(obj):
            errors.append(f"{path}: Invalid value: \"{obj}\": not RFC3339")
        return

    if stype == "integer":
        if isinstance(obj, bool) or not isinstance(obj, int):
            errors.append(f"{path}: expected integer, got {type(obj).__name__}")
            return
        if "minimum" in schema and obj < schema["minimum"]:
            errors.append(
                f"{path}: Invalid value: {obj}: must be greater than or "
                f"equal to {schema['minimum']}"
            )
        if "maximum" in schema and obj > schema["maximum"]:
            errors.append(f"{path}: above maximum {schema['maximum']}")
        return

    if stype == "number":
        if isinstance(obj, bool) or not isinstance(obj, (int, float)):
            errors.append(f"{path}: expected number, got {type(obj).__name__}")
        return

    if stype == "boolean":
        if not isinstance(obj, bool):
            errors.append(f"{path}: expected boolean, got {type(obj).__name__}")
        return
    # no/unknown type: accept


def merge_patch(target: Any, patch: Any) -> Any:
    """RFC 7386 JSON merge patch (what Content-Type
    application/merge-patch+json means to the API server)."""
    if not isinstance(patch, dict):
        return patch
    if not isinstance(target, dict):
        target = {}
    out = dict(target)
    for key, val in patch.items():
        if val is None:
            out.pop(key, None)
        else:
            out[key] = merge_patch(out.get(key), val)
    return out
