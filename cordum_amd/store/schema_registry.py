"""JSON-Schema registry + validator.

Oracle: core/infra/schema/registry.go:19-160 + validate.go:11-61 — schemas
stored by id (cap 500), validation used for workflow input and step in/out
payloads and for config-file validation.

The validator is a self-contained JSON-Schema (draft-07 subset) implementation
covering the keywords the reference's workflows/packs use: type, properties,
required, items (incl. tuple form + additionalItems), enum, const,
minimum/maximum (+exclusive), multipleOf, minLength/maxLength, pattern,
additionalProperties, min/maxProperties, min/maxItems, uniqueItems,
anyOf/allOf/oneOf/not, local $ref (#/definitions, #/$defs), format (opaque).
Conformance corpus: tests/test_schema_conformance.py (JSON-Schema-Test-Suite
style triples).
"""
from __future__ import annotations

import re
import threading
from typing import Any, Dict, List, Optional, Tuple

SCHEMA_CAP = 500


class SchemaValidationError(Exception):
    def __init__(self, errors: List[str]):
        super().__init__("; ".join(errors))
        self.errors = errors


def _type_ok(t: str, v: Any) -> bool:
    if t == "object":
        return isinstance(v, dict)
    if t == "array":
        return isinstance(v, list)
    if t == "string":
        return isinstance(v, str)
    if t == "number":
        return isinstance(v, (int, float)) and not isinstance(v, bool)
    if t == "integer":
        return isinstance(v, int) and not isinstance(v, bool)
    if t == "boolean":
        return isinstance(v, bool)
    if t == "null":
        return v is None
    return True


def validate_value(schema: Dict[str, Any], value: Any, path: str = "$",
                   root: Optional[Dict[str, Any]] = None) -> List[str]:
    errs: List[str] = []
    if not isinstance(schema, dict):
        return errs
    if root is None:
        root = schema
    # local $ref resolution (#/definitions/... or #/$defs/...), draft-07
    ref = schema.get("$ref")
    if isinstance(ref, str) and ref.startswith("#/"):
        target: Any = root
        for part in ref[2:].split("/"):
            part = part.replace("~1", "/").replace("~0", "~")
            if isinstance(target, dict) and part in target:
                target = target[part]
            else:
                return [f"{path}: unresolvable $ref {ref}"]
        return validate_value(target, value, path, root)
    t = schema.get("type")
    if t is not None:
        types = t if isinstance(t, list) else [t]
        if not any(_type_ok(x, value) for x in types):
            errs.append(f"{path}: expected {t}, got {type(value).__name__}")
            return errs
    if "enum" in schema and value not in schema["enum"]:
        errs.append(f"{path}: not in enum {schema['enum']}")
    if "const" in schema and value != schema["const"]:
        errs.append(f"{path}: != const {schema['const']}")
    if isinstance(value, (int, float)) and not isinstance(value, bool):
        if "minimum" in schema and value < schema["minimum"]:
            errs.append(f"{path}: {value} < minimum {schema['minimum']}")
        if "maximum" in schema and value > schema["maximum"]:
            errs.append(f"{path}: {value} > maximum {schema['maximum']}")
        if "exclusiveMinimum" in schema and value <= schema["exclusiveMinimum"]:
            errs.append(f"{path}: {value} <= exclusiveMinimum {schema['exclusiveMinimum']}")
        if "exclusiveMaximum" in schema and value >= schema["exclusiveMaximum"]:
            errs.append(f"{path}: {value} >= exclusiveMaximum {schema['exclusiveMaximum']}")
        if "multipleOf" in schema:
            m = schema["multipleOf"]
            q = value / m
            if abs(q - round(q)) > 1e-9:
                errs.append(f"{path}: {value} not a multiple of {m}")
    if isinstance(value, str):
        if "minLength" in schema and len(value) < schema["minLength"]:
            errs.append(f"{path}: shorter than minLength {schema['minLength']}")
        if "maxLength" in schema and len(value) > schema["maxLength"]:
            errs.append(f"{path}: longer than maxLength {schema['maxLength']}")
        if "pattern" in schema:
            try:
                if not re.search(schema["pattern"], value):
                    errs.append(f"{path}: does not match pattern {schema['pattern']}")
            except re.error:
                pass
    if isinstance(value, dict):
        props = schema.get("properties", {})
        for req in schema.get("required", []):
            if req not in value:
                errs.append(f"{path}: missing required property {req!r}")
        if "minProperties" in schema and len(value) < schema["minProperties"]:
            errs.append(f"{path}: fewer than minProperties {schema['minProperties']}")
        if "maxProperties" in schema and len(value) > schema["maxProperties"]:
            errs.append(f"{path}: more than maxProperties {schema['maxProperties']}")
        for k, v in value.items():
            if k in props:
                errs.extend(validate_value(props[k], v, f"{path}.{k}", root))
            elif schema.get("additionalProperties") is False:
                errs.append(f"{path}: unexpected property {k!r}")
            elif isinstance(schema.get("additionalProperties"), dict):
                errs.extend(validate_value(schema["additionalProperties"], v, f"{path}.{k}", root))
    if isinstance(value, list):
        items = schema.get("items")
        if isinstance(items, dict):
            for i, item in enumerate(value):
                errs.extend(validate_value(items, item, f"{path}[{i}]", root))
        elif isinstance(items, list):  # draft-07 tuple validation
            for i, sub in enumerate(items):
                if i < len(value):
                    errs.extend(validate_value(sub, value[i], f"{path}[{i}]", root))
            extra = schema.get("additionalItems")
            if extra is False and len(value) > len(items):
                errs.append(f"{path}: more items than tuple schema allows")
            elif isinstance(extra, dict):
                for i in range(len(items), len(value)):
                    errs.extend(validate_value(extra, value[i], f"{path}[{i}]", root))
        if "minItems" in schema and len(value) < schema["minItems"]:
            errs.append(f"{path}: fewer than minItems {schema['minItems']}")
        if "maxItems" in schema and len(value) > schema["maxItems"]:
            errs.append(f"{path}: more than maxItems {schema['maxItems']}")
        if schema.get("uniqueItems"):
            seen = []
            for item in value:
                if item in seen:
                    errs.append(f"{path}: items not unique")
                    break
                seen.append(item)
    if "not" in schema:
        if not validate_value(schema["not"], value, path, root):
            errs.append(f"{path}: matches schema in 'not'")
    for comb, mode in (("anyOf", "any"), ("allOf", "all"), ("oneOf", "one")):
        if comb in schema:
            sub_errs = [validate_value(s, value, path, root) for s in schema[comb]]
            ok = [not e for e in sub_errs]
            if mode == "any" and not any(ok):
                errs.append(f"{path}: no {comb} branch matched")
            elif mode == "all" and not all(ok):
                errs.extend(e for es in sub_errs for e in es)
            elif mode == "one" and sum(ok) != 1:
                errs.append(f"{path}: {sum(ok)} oneOf branches matched (want 1)")
    return errs


def validate(schema: Dict[str, Any], value: Any) -> None:
    errs = validate_value(schema, value)
    if errs:
        raise SchemaValidationError(errs)


class SchemaRegistry:
    def __init__(self, cap: int = SCHEMA_CAP):
        self._mu = threading.Lock()
        self._schemas: Dict[str, Dict[str, Any]] = {}
        self._order: List[str] = []
        self._cap = cap

    def put(self, schema_id: str, schema: Dict[str, Any]) -> None:
        if not schema_id:
            raise ValueError("schema id required")
        if not isinstance(schema, dict):
            raise ValueError("schema must be an object")
        with self._mu:
            if schema_id not in self._schemas:
                self._order.append(schema_id)
                while len(self._order) > self._cap:
                    victim = self._order.pop(0)
                    self._schemas.pop(victim, None)
            self._schemas[schema_id] = schema

    def get(self, schema_id: str) -> Optional[Dict[str, Any]]:
        with self._mu:
            return self._schemas.get(schema_id)

    def delete(self, schema_id: str) -> bool:
        with self._mu:
            if schema_id in self._schemas:
                del self._schemas[schema_id]
                self._order.remove(schema_id)
                return True
            return False

    def list(self) -> List[str]:
        with self._mu:
            return list(self._order)

    def validate_against(self, schema_id: str, value: Any) -> Tuple[bool, List[str]]:
        schema = self.get(schema_id)
        if schema is None:
            return False, [f"schema {schema_id!r} not found"]
        errs = validate_value(schema, value)
        return not errs, errs
