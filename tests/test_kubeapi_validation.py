"""openAPIV3Schema validator edges of the apiserver stand-in (the subset our
CRD uses: type/required/properties/items/minLength/minItems/enum/minimum)."""
import pytest

from inferno_amd.testing.kubeapi import ValidationError, validate_schema


class TestValidateSchema:
    def test_required_nested(self):
        schema = {"type": "object", "properties": {
            "spec": {"type": "object", "required": ["modelID"],
                     "properties": {"modelID": {"type": "string"}}}}}
        validate_schema({"spec": {"modelID": "m"}}, schema)
        with pytest.raises(ValidationError, match="modelID"):
            validate_schema({"spec": {}}, schema)

    def test_type_mismatches(self):
        with pytest.raises(ValidationError):
            validate_schema("x", {"type": "object"})
        with pytest.raises(ValidationError):
            validate_schema({"a": 1}, {"type": "object",
                                       "properties": {"a": {"type": "string"}}})
        with pytest.raises(ValidationError):
            validate_schema(1.5, {"type": "integer"})
        # bool is NOT an integer (k8s structural-schema semantics)
        with pytest.raises(ValidationError):
            validate_schema(True, {"type": "integer"})
        validate_schema(3, {"type": "integer"})
        validate_schema(3, {"type": "number"})
        validate_schema(True, {"type": "boolean"})

    def test_min_constraints(self):
        with pytest.raises(ValidationError):
            validate_schema("", {"type": "string", "minLength": 1})
        with pytest.raises(ValidationError):
            validate_schema([], {"type": "array", "minItems": 1,
                                 "items": {"type": "string"}})
        with pytest.raises(ValidationError):
            validate_schema(0, {"type": "integer", "minimum": 1})
        validate_schema(1, {"type": "integer", "minimum": 1})

    def test_enum(self):
        s = {"type": "string", "enum": ["A", "B"]}
        validate_schema("A", s)
        with pytest.raises(ValidationError):
            validate_schema("C", s)

    def test_items_recursion_and_paths(self):
        schema = {"type": "array", "items": {
            "type": "object", "required": ["acc"],
            "properties": {"acc": {"type": "string", "minLength": 1}}}}
        validate_schema([{"acc": "MI355X"}], schema)
        with pytest.raises(ValidationError, match=r"\[1\]"):
            validate_schema([{"acc": "x"}, {"nope": 1}], schema)

    def test_unknown_fields_tolerated(self):
        # structural-schema default: unknown fields pruned-tolerant
        validate_schema({"a": 1, "extra": {"deep": True}},
                        {"type": "object", "properties": {"a": {"type": "integer"}}})

    def test_untyped_accepts_anything(self):
        validate_schema({"x": [1, "y"]}, {})
