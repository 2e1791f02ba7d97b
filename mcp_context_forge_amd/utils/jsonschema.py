"""Compact JSON-Schema validator (draft-07 subset).

Reference analog: plugins/schema_guard/schema_guard.py (jsonschema pip dep)
and tool output_schema checks in tool_service. Supports: type, properties,
required, additionalProperties, items, enum, const, minimum/maximum,
exclusiveMinimum/Maximum, minLength/maxLength, pattern, minItems/maxItems,
anyOf/allOf/oneOf, nullable via type lists.
"""

from __future__ import annotations

import re
from typing import Any, List


class SchemaError(Exception):
    def __init__(self, path: str, message: str):
        self.path = path
        self.message = message
        super().__init__(f"{path or '$'}: {message}")


_TYPES = {
    "object": dict,
    "array": list,
    "string": str,
    "integer": int,
    "number": (int, float),
    "boolean": bool,
    "null": type(None),
}


def _check_type(value: Any, ty: str) -> bool:
    py = _TYPES.get(ty)
    if py is None:
        return True
    if ty == "integer" and isinstance(value, bool):
        return False
    if ty == "number" and isinstance(value, bool):
        return False
    return isinstance(value, py)


def validate(value: Any, schema: Any, path: str = "$") -> List[str]:
    """Returns a list of violation messages (empty = valid)."""
    errs: List[str] = []
    if not isinstance(schema, dict):
        return errs

    ty = schema.get("type")
    if ty is not None:
        types = ty if isinstance(ty, list) else [ty]
        if not any(_check_type(value, t) for t in types):
            errs.append(f"{path}: expected type {ty}, got {type(value).__name__}")
            return errs

    if "enum" in schema and value not in schema["enum"]:
        errs.append(f"{path}: {value!r} not in enum")
    if "const" in schema and value != schema["const"]:
        errs.append(f"{path}: must equal const")

    if isinstance(value, (int, float)) and not isinstance(value, bool):
        if "minimum" in schema and value < schema["minimum"]:
            errs.append(f"{path}: {value} < minimum {schema['minimum']}")
        if "maximum" in schema and value > schema["maximum"]:
            errs.append(f"{path}: {value} > maximum {schema['maximum']}")
        if "exclusiveMinimum" in schema and value <= schema["exclusiveMinimum"]:
            errs.append(f"{path}: {value} <= exclusiveMinimum")
        if "exclusiveMaximum" in schema and value >= schema["exclusiveMaximum"]:
            errs.append(f"{path}: {value} >= exclusiveMaximum")

    if isinstance(value, str):
        if "minLength" in schema and len(value) < schema["minLength"]:
            errs.append(f"{path}: shorter than minLength {schema['minLength']}")
        if "maxLength" in schema and len(value) > schema["maxLength"]:
            errs.append(f"{path}: longer than maxLength {schema['maxLength']}")
        if "pattern" in schema and not re.search(schema["pattern"], value):
            errs.append(f"{path}: does not match pattern")

    if isinstance(value, dict):
        props = schema.get("properties", {})
        for req in schema.get("required", []):
            if req not in value:
                errs.append(f"{path}: missing required property {req!r}")
        for k, v in value.items():
            if k in props:
                errs.extend(validate(v, props[k], f"{path}.{k}"))
            elif schema.get("additionalProperties") is False:
                errs.append(f"{path}: additional property {k!r} not allowed")
            elif isinstance(schema.get("additionalProperties"), dict):
                errs.extend(validate(v, schema["additionalProperties"], f"{path}.{k}"))

    if isinstance(value, list):
        if "minItems" in schema and len(value) < schema["minItems"]:
            errs.append(f"{path}: fewer than minItems")
        if "maxItems" in schema and len(value) > schema["maxItems"]:
            errs.append(f"{path}: more than maxItems")
        items = schema.get("items")
        if isinstance(items, dict):
            for i, item in enumerate(value):
                errs.extend(validate(item, items, f"{path}[{i}]"))

    for comb, need in (("allOf", "all"), ("anyOf", "any"), ("oneOf", "one")):
        subs = schema.get(comb)
        if subs:
            results = [validate(value, s, path) for s in subs]
            ok = [not r for r in results]
            if need == "all" and not all(ok):
                for r in results:
                    errs.extend(r)
            elif need == "any" and not any(ok):
                errs.append(f"{path}: does not match anyOf")
            elif need == "one" and sum(ok) != 1:
                errs.append(f"{path}: must match exactly one of oneOf")
    return errs


def is_valid(value: Any, schema: Any) -> bool:
    return not validate(value, schema)
