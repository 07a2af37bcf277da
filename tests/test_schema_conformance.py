"""Conformance corpus for the self-contained JSON-Schema validator
(store/schema_registry.py — the reference validates with
santhosh-tekuri/jsonschema/v5, registry.go:19-160 + validate.go:11-61).

Cases are authored in the official JSON-Schema-Test-Suite style:
(schema, instance, valid?) triples per keyword, covering the draft-07
subset pack schemas and workflow step I/O schemas actually use."""
import pytest

from cordum_amd.store.schema_registry import validate_value

CASES = [
    # --- type ---------------------------------------------------------------
    ("type-int-ok", {"type": "integer"}, 3, True),
    ("type-int-bool-is-not-int", {"type": "integer"}, True, False),
    ("type-num-accepts-int", {"type": "number"}, 3, True),
    ("type-num-bool-rejected", {"type": "number"}, True, False),
    ("type-str", {"type": "string"}, "x", True),
    ("type-str-rejects-num", {"type": "string"}, 3, False),
    ("type-null", {"type": "null"}, None, True),
    ("type-union", {"type": ["string", "null"]}, None, True),
    ("type-union-miss", {"type": ["string", "null"]}, 3, False),
    ("type-object", {"type": "object"}, {}, True),
    ("type-array", {"type": "array"}, [], True),
    # --- enum / const -------------------------------------------------------
    ("enum-ok", {"enum": [1, "a", None]}, "a", True),
    ("enum-miss", {"enum": [1, "a"]}, "b", False),
    ("const-ok", {"const": 5}, 5, True),
    ("const-miss", {"const": 5}, 6, False),
    # --- numeric ------------------------------------------------------------
    ("min-ok", {"minimum": 2}, 2, True),
    ("min-miss", {"minimum": 2}, 1.5, False),
    ("max-ok", {"maximum": 2}, 2, True),
    ("max-miss", {"maximum": 2}, 2.5, False),
    ("exclmin-ok", {"exclusiveMinimum": 2}, 2.1, True),
    ("exclmin-miss", {"exclusiveMinimum": 2}, 2, False),
    ("exclmax-ok", {"exclusiveMaximum": 2}, 1.9, True),
    ("exclmax-miss", {"exclusiveMaximum": 2}, 2, False),
    ("multipleof-ok", {"multipleOf": 0.5}, 2.5, True),
    ("multipleof-miss", {"multipleOf": 2}, 3, False),
    # --- string -------------------------------------------------------------
    ("minlen-ok", {"minLength": 2}, "ab", True),
    ("minlen-miss", {"minLength": 2}, "a", False),
    ("maxlen-miss", {"maxLength": 2}, "abc", False),
    ("pattern-ok", {"pattern": "^job\\."}, "job.echo", True),
    ("pattern-miss", {"pattern": "^job\\."}, "sys.echo", False),
    # --- object -------------------------------------------------------------
    ("required-ok", {"required": ["a"]}, {"a": 1}, True),
    ("required-miss", {"required": ["a", "b"]}, {"a": 1}, False),
    ("props-nested", {"properties": {"a": {"type": "integer"}}}, {"a": "x"}, False),
    ("addprops-false-ok", {"properties": {"a": {}}, "additionalProperties": False},
     {"a": 1}, True),
    ("addprops-false-miss", {"properties": {"a": {}}, "additionalProperties": False},
     {"a": 1, "b": 2}, False),
    ("addprops-schema", {"additionalProperties": {"type": "string"}},
     {"x": "ok", "y": 3}, False),
    ("minprops-miss", {"minProperties": 2}, {"a": 1}, False),
    ("maxprops-miss", {"maxProperties": 1}, {"a": 1, "b": 2}, False),
    # --- array --------------------------------------------------------------
    ("items-ok", {"items": {"type": "integer"}}, [1, 2, 3], True),
    ("items-miss", {"items": {"type": "integer"}}, [1, "x"], False),
    ("tuple-ok", {"items": [{"type": "integer"}, {"type": "string"}]},
     [1, "a"], True),
    ("tuple-miss", {"items": [{"type": "integer"}, {"type": "string"}]},
     ["a", 1], False),
    ("tuple-additional-false", {"items": [{"type": "integer"}],
                                "additionalItems": False}, [1, 2], False),
    ("tuple-additional-schema", {"items": [{}], "additionalItems": {"type": "integer"}},
     [None, "x"], False),
    ("minitems-miss", {"minItems": 2}, [1], False),
    ("maxitems-miss", {"maxItems": 1}, [1, 2], False),
    ("unique-ok", {"uniqueItems": True}, [1, 2, [3]], True),
    ("unique-miss", {"uniqueItems": True}, [1, 2, 1], False),
    ("unique-deep-miss", {"uniqueItems": True}, [{"a": 1}, {"a": 1}], False),
    # --- combinators --------------------------------------------------------
    ("anyof-ok", {"anyOf": [{"type": "string"}, {"type": "integer"}]}, 3, True),
    ("anyof-miss", {"anyOf": [{"type": "string"}, {"type": "integer"}]}, 1.5, False),
    ("allof-ok", {"allOf": [{"minimum": 1}, {"maximum": 3}]}, 2, True),
    ("allof-miss", {"allOf": [{"minimum": 1}, {"maximum": 3}]}, 5, False),
    ("oneof-ok", {"oneOf": [{"type": "integer"}, {"minimum": 10}]}, 5, True),
    ("oneof-both-match", {"oneOf": [{"type": "integer"}, {"minimum": 1}]}, 5, False),
    ("not-ok", {"not": {"type": "string"}}, 5, True),
    ("not-miss", {"not": {"type": "string"}}, "x", False),
    # --- $ref ---------------------------------------------------------------
    ("ref-definitions-ok",
     {"definitions": {"pos": {"type": "integer", "minimum": 1}},
      "properties": {"n": {"$ref": "#/definitions/pos"}}},
     {"n": 5}, True),
    ("ref-definitions-miss",
     {"definitions": {"pos": {"type": "integer", "minimum": 1}},
      "properties": {"n": {"$ref": "#/definitions/pos"}}},
     {"n": 0}, False),
    ("ref-defs-ok",
     {"$defs": {"s": {"type": "string"}},
      "items": {"$ref": "#/$defs/s"}}, ["a", "b"], True),
    ("ref-unresolvable",
     {"properties": {"n": {"$ref": "#/definitions/nope"}}}, {"n": 1}, False),
    # --- realistic pack/workflow shapes -------------------------------------
    ("echo-input-ok",
     {"type": "object", "required": ["message"],
      "properties": {"message": {"type": "string"},
                     "author": {"type": "string"}},
      "additionalProperties": False},
     {"message": "hi", "author": "a"}, True),
    ("echo-input-miss",
     {"type": "object", "required": ["message"],
      "properties": {"message": {"type": "string"}},
      "additionalProperties": False},
     {"author": "a"}, False),
    ("fanout-items",
     {"type": "object",
      "properties": {"items": {"type": "array", "minItems": 1,
                               "items": {"type": "object",
                                         "required": ["id"]}}}},
     {"items": [{"id": 1}, {}]}, False),
]


@pytest.mark.parametrize("name,schema,value,want_valid", CASES,
                         ids=[c[0] for c in CASES])
def test_schema_conformance(name, schema, value, want_valid):
    errs = validate_value(schema, value)
    assert (not errs) == want_valid, errs
