"""Config file resolution, loading, and strict validation.

Mirrors the reference's CLI config contract (reference
shipyard.py:441-580 `CliContext._init_config`, convoy/validator.py):
eight YAML families resolved from a --configdir or per-file paths, each
strictly validated against its schema before any action.
"""
from __future__ import annotations

import enum
import os
from pathlib import Path
from typing import Any, Dict, Optional

import yaml

from shipyard_amd.config.schema import SchemaViolation, Validator

_SCHEMA_DIR = Path(__file__).resolve().parent / "schemas"


class ConfigType(enum.Enum):
    credentials = "credentials"
    config = "config"
    pool = "pool"
    jobs = "jobs"
    fs = "fs"
    monitor = "monitor"
    federation = "federation"
    slurm = "slurm"


_validators: Dict[ConfigType, Validator] = {}


def get_validator(ctype: ConfigType) -> Validator:
    v = _validators.get(ctype)
    if v is None:
        with open(_SCHEMA_DIR / f"{ctype.value}.yaml") as f:
            v = Validator(yaml.safe_load(f))
        _validators[ctype] = v
    return v


def validate_config(ctype: ConfigType, doc: Any, name: str = "") -> None:
    """Strict validation (reference convoy/validator.py:112-125)."""
    get_validator(ctype).validate(doc, name or ctype.value)


def load_config_file(path, ctype: Optional[ConfigType] = None,
                     validate: bool = True) -> Dict[str, Any]:
    with open(path) as f:
        doc = yaml.safe_load(f) or {}
    if validate and ctype is not None:
        validate_config(ctype, doc, str(path))
    return doc


_FAMILY_FILENAMES = {
    ConfigType.credentials: "credentials.yaml",
    ConfigType.config: "config.yaml",
    ConfigType.pool: "pool.yaml",
    ConfigType.jobs: "jobs.yaml",
    ConfigType.fs: "fs.yaml",
    ConfigType.monitor: "monitor.yaml",
    ConfigType.federation: "federation.yaml",
    ConfigType.slurm: "slurm.yaml",
}


def resolve_config_files(configdir: Optional[str] = None,
                         **overrides) -> Dict[ConfigType, Path]:
    """--configdir / per-file flags with env fallbacks
    (reference shipyard.py:804-871).  overrides: credentials=..., etc."""
    out: Dict[ConfigType, Path] = {}
    for ctype, fname in _FAMILY_FILENAMES.items():
        cand = overrides.get(ctype.value)
        if cand is None:
            cand = os.environ.get(f"SHIPYARD_{ctype.value.upper()}_CONF")
        if cand is None and configdir:
            p = Path(configdir) / fname
            if p.exists():
                cand = p
            else:
                # also accept .yml
                alt = Path(configdir) / fname.replace(".yaml", ".yml")
                if alt.exists():
                    cand = alt
        if cand is not None:
            out[ctype] = Path(cand)
    return out


class ConfigBundle:
    """All loaded config families for one invocation."""

    def __init__(self, docs: Dict[ConfigType, Dict[str, Any]]):
        self.docs = docs

    @classmethod
    def from_dir(cls, configdir: str, validate: bool = True,
                 **overrides) -> "ConfigBundle":
        files = resolve_config_files(configdir, **overrides)
        docs = {}
        for ctype, path in files.items():
            docs[ctype] = load_config_file(path, ctype, validate=validate)
        return cls(docs)

    def get(self, ctype: ConfigType) -> Optional[Dict[str, Any]]:
        return self.docs.get(ctype)

    def require(self, ctype: ConfigType) -> Dict[str, Any]:
        doc = self.docs.get(ctype)
        if doc is None:
            raise KeyError(f"config family not loaded: {ctype.value}")
        return doc


__all__ = ["ConfigType", "ConfigBundle", "SchemaViolation",
           "load_config_file", "resolve_config_files", "validate_config"]
