"""Direct unit tests of the strict schema engine
(shipyard_amd/config/schema.py) — the pykwalify-analogue's own rule
semantics, independent of the shipped family schemas."""
import pytest

from shipyard_amd.config.schema import SchemaViolation, Validator


def check(schema, doc, ok=True):
    v = Validator(schema)
    if ok:
        v.validate(doc, "t")
    else:
        with pytest.raises(SchemaViolation):
            v.validate(doc, "t")


def test_required_vs_nullable():
    schema = {"type": "map", "mapping": {
        "a": {"type": "str", "required": True},
        "b": {"type": "str", "nullable": False},
    }}
    check(schema, {"a": "x"})
    check(schema, {}, ok=False)             # a missing
    check(schema, {"a": None}, ok=False)    # a null
    check(schema, {"a": "x", "b": None}, ok=False)  # b not nullable


def test_bool_is_not_int():
    schema = {"type": "map", "mapping": {"n": {"type": "int"}}}
    check(schema, {"n": 3})
    check(schema, {"n": True}, ok=False)


def test_text_accepts_str_and_number():
    schema = {"type": "map", "mapping": {"v": {"type": "text"}}}
    check(schema, {"v": "all"})
    check(schema, {"v": 4})
    check(schema, {"v": True}, ok=False)
    check(schema, {"v": ["x"]}, ok=False)


def test_pattern_fullmatch():
    schema = {"type": "map", "mapping": {
        "id": {"type": "str", "pattern": "[a-z]+"}}}
    check(schema, {"id": "abc"})
    check(schema, {"id": "abc!"}, ok=False)
    check(schema, {"id": "Abc"}, ok=False)


def test_range_on_numbers_and_lengths():
    schema = {"type": "map", "mapping": {
        "n": {"type": "int", "range": {"min": 1, "max": 8}},
        "s": {"type": "str", "range": {"min": 2}},
        "l": {"type": "seq", "range": {"max": 2},
              "sequence": [{"type": "int"}]},
    }}
    check(schema, {"n": 8, "s": "ab", "l": [1, 2]})
    check(schema, {"n": 0}, ok=False)
    check(schema, {"s": "a"}, ok=False)
    check(schema, {"l": [1, 2, 3]}, ok=False)


def test_wildcard_mapping_and_strictness():
    schema = {"type": "map", "mapping": {
        "known": {"type": "int"},
        "=": {"type": "str"},
    }}
    check(schema, {"known": 1, "anything": "str-ok"})
    check(schema, {"anything": 5}, ok=False)  # wildcard enforces type
    strict = {"type": "map", "mapping": {"known": {"type": "int"}}}
    check(strict, {"unknown": 1}, ok=False)
    loose = {"type": "map", "allow_unknown": True,
             "mapping": {"known": {"type": "int"}}}
    check(loose, {"unknown": 1})


def test_nested_sequence_paths_in_errors():
    schema = {"type": "map", "mapping": {
        "items": {"type": "seq", "sequence": [
            {"type": "map", "mapping": {
                "x": {"type": "int", "required": True}}}]}}}
    v = Validator(schema)
    with pytest.raises(SchemaViolation) as ei:
        v.validate({"items": [{"x": 1}, {}]}, "doc")
    assert "doc.items[1].x" in str(ei.value)


def test_timedelta_and_size_rules():
    schema = {"type": "map", "mapping": {
        "t": {"type": "timedelta"}, "z": {"type": "size"}}}
    check(schema, {"t": "1.02:03:04", "z": "256mi"})
    check(schema, {"t": "tomorrow"}, ok=False)
    check(schema, {"z": "many bytes"}, ok=False)
    check(schema, {"z": 1024})


def test_enum():
    schema = {"type": "map", "mapping": {
        "mode": {"type": "str", "enum": ["pack", "spread"]}}}
    check(schema, {"mode": "pack"})
    check(schema, {"mode": "stack"}, ok=False)
