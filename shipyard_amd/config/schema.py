"""Strict YAML schema validation engine.

The reference validates every config file against pykwalify schemas with
``strict_rule_validation`` before any action (reference
convoy/validator.py:112-125).  pykwalify is not in this image, so this is
a small, purpose-built engine with the semantics the framework needs:

rule keys:
  type:       str | int | float | bool | number | text | map | seq | any
              | timedelta | size
  required:   bool (default False)
  nullable:   bool (default True when not required, schemas may override)
  enum:       list of allowed scalars
  pattern:    regex (fullmatch) for str values
  range:      {min, max} for numbers / string length
  mapping:    {key: rule} for maps; unknown keys are ERRORS unless
              'allow_unknown: true' or a '=' wildcard rule is present
  sequence:   [rule] — rule applied to each element
  desc:       documentation only

'text' accepts str or number (pykwalify semantics).  'timedelta' accepts
'[d.]HH:MM:SS' strings; 'size' accepts ints or '256m'-style strings.
Errors carry the full config path for actionable messages.
"""
from __future__ import annotations

import re
from typing import Any, Dict, List

from shipyard_amd import utils


class SchemaViolation(ValueError):
    def __init__(self, errors: List[str]):
        self.errors = errors
        super().__init__("config validation failed:\n  " +
                         "\n  ".join(errors))


_SCALARS = {
    "str": (str,),
    "int": (int,),
    "float": (float, int),
    "bool": (bool,),
    "number": (int, float),
}


def _type_ok(rule_type: str, value: Any) -> bool:
    if rule_type == "any":
        return True
    if rule_type == "text":
        return isinstance(value, (str, int, float)) and not isinstance(
            value, bool)
    if rule_type == "timedelta":
        if not isinstance(value, str):
            return False
        try:
            utils.parse_timedelta(value)
            return True
        except ValueError:
            return False
    if rule_type == "size":
        if isinstance(value, bool):
            return False
        if isinstance(value, int):
            return True
        if isinstance(value, str):
            try:
                utils.parse_size(value)
                return True
            except ValueError:
                return False
        return False
    if rule_type in ("int", "number", "float"):
        if isinstance(value, bool):
            return False
        return isinstance(value, _SCALARS[rule_type])
    if rule_type in _SCALARS:
        return isinstance(value, _SCALARS[rule_type])
    if rule_type == "map":
        return isinstance(value, dict)
    if rule_type == "seq":
        return isinstance(value, list)
    raise ValueError(f"unknown schema type: {rule_type}")


class Validator:
    def __init__(self, schema: Dict[str, Any]):
        self.schema = schema

    def validate(self, doc: Any, doc_name: str = "config") -> None:
        errors: List[str] = []
        self._walk(self.schema, doc, doc_name, errors)
        if errors:
            raise SchemaViolation(errors)

    # -- internals ---------------------------------------------------
    def _walk(self, rule: Dict[str, Any], value: Any, path: str,
              errors: List[str]) -> None:
        if not isinstance(rule, dict):
            errors.append(f"{path}: schema rule is not a map")
            return
        rtype = rule.get("type", "any")

        if value is None:
            if rule.get("required", False) and not rule.get("nullable",
                                                            False):
                errors.append(f"{path}: required value is missing/null")
            elif not rule.get("nullable", True):
                errors.append(f"{path}: value may not be null")
            return

        if not _type_ok(rtype, value):
            errors.append(
                f"{path}: expected {rtype}, got {type(value).__name__} "
                f"({value!r})")
            return

        if "enum" in rule and value not in rule["enum"]:
            errors.append(f"{path}: {value!r} not in {rule['enum']}")
        if "pattern" in rule and isinstance(value, str):
            if not re.fullmatch(rule["pattern"], value):
                errors.append(
                    f"{path}: {value!r} does not match /{rule['pattern']}/")
        if "range" in rule:
            r = rule["range"]
            mag = len(value) if isinstance(value, (str, list, dict)) else value
            if "min" in r and r["min"] is not None and mag < r["min"]:
                errors.append(f"{path}: {mag} < min {r['min']}")
            if "max" in r and r["max"] is not None and mag > r["max"]:
                errors.append(f"{path}: {mag} > max {r['max']}")

        if rtype == "map":
            mapping = rule.get("mapping")
            if mapping is not None:
                wildcard = mapping.get("=")
                for key, sub in mapping.items():
                    if key == "=":
                        continue
                    if key in value:
                        self._walk(sub, value[key], f"{path}.{key}", errors)
                    elif sub.get("required", False):
                        errors.append(f"{path}.{key}: required key missing")
                for key in value:
                    if key not in mapping:
                        if wildcard is not None:
                            self._walk(wildcard, value[key],
                                       f"{path}.{key}", errors)
                        elif not rule.get("allow_unknown", False):
                            errors.append(f"{path}.{key}: unknown key")
        elif rtype == "seq":
            seq_rule = rule.get("sequence")
            if seq_rule:
                elem_rule = seq_rule[0] if isinstance(seq_rule, list) \
                    else seq_rule
                for i, item in enumerate(value):
                    self._walk(elem_rule, item, f"{path}[{i}]", errors)
