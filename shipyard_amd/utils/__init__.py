"""Shared utilities (analogue of reference convoy/util.py).

Logging setup, base64/hash helpers, timedelta parsing, shell command
wrapping and subprocess execution — re-implemented for the local
MI355X executor (no Azure, no Python 2 compat).
"""
from __future__ import annotations

import base64
import datetime
import hashlib
import logging
import os
import re
import subprocess
import sys
from typing import List, Optional, Sequence

_LOG_FORMAT = "%(asctime)s %(levelname)s %(name)s:%(funcName)s:%(lineno)d %(message)s"


def setup_logger(logger: logging.Logger, level: int = logging.INFO) -> None:
    """Attach the shared formatter (reference convoy/util.py:86-114)."""
    if logger.handlers:
        return
    handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(logging.Formatter(_LOG_FORMAT))
    logger.addHandler(handler)
    logger.setLevel(level)


def get_logger(name: str) -> logging.Logger:
    logger = logging.getLogger(name)
    setup_logger(logger)
    return logger


def base64_encode_string(s: str) -> str:
    return base64.b64encode(s.encode("utf8")).decode("ascii")


def base64_decode_string(s: str) -> str:
    return base64.b64decode(s).decode("utf8")


def hash_string(s: str) -> str:
    return hashlib.sha1(s.encode("utf8")).hexdigest()


_TIMEDELTA_RE = re.compile(
    r"^(?:(?P<days>\d+)\.)?(?P<hours>\d+):(?P<minutes>\d+):(?P<seconds>\d+)$")


def parse_timedelta(value) -> Optional[datetime.timedelta]:
    """Parse '[d.]HH:MM:SS' (the reference's duration format,
    convoy/util.py) into a timedelta; passthrough for timedelta/None."""
    if value is None:
        return None
    if isinstance(value, datetime.timedelta):
        return value
    m = _TIMEDELTA_RE.match(str(value).strip())
    if not m:
        raise ValueError(f"invalid timedelta: {value!r} (want [d.]HH:MM:SS)")
    parts = {k: int(v) for k, v in m.groupdict(default="0").items()}
    return datetime.timedelta(**parts)


_SIZE_RE = re.compile(r"^(?P<num>\d+(?:\.\d+)?)\s*(?P<unit>[kmgt]?i?b?)$",
                      re.IGNORECASE)
_SIZE_MULT = {
    "": 1, "b": 1,
    "k": 10 ** 3, "kb": 10 ** 3, "ki": 2 ** 10, "kib": 2 ** 10,
    "m": 10 ** 6, "mb": 10 ** 6, "mi": 2 ** 20, "mib": 2 ** 20,
    "g": 10 ** 9, "gb": 10 ** 9, "gi": 2 ** 30, "gib": 2 ** 30,
    "t": 10 ** 12, "tb": 10 ** 12, "ti": 2 ** 40, "tib": 2 ** 40,
}


def parse_size(value) -> Optional[int]:
    """Parse '256m' / '4gi' style sizes into bytes (docker --shm-size
    convention: k/m/g are decimal, ki/mi/gi binary)."""
    if value is None:
        return None
    if isinstance(value, int):
        return value
    m = _SIZE_RE.match(str(value).strip())
    if not m:
        raise ValueError(f"invalid size: {value!r}")
    return int(float(m.group("num")) * _SIZE_MULT[m.group("unit").lower()])


def wrap_commands_in_shell(commands: Sequence[str], wait: bool = True) -> str:
    """Single /bin/bash -c line from a command list (reference
    convoy/util.py:368 `wrap_commands_in_shell`)."""
    joined = "; ".join(commands)
    return f"/bin/bash -c 'set -e; set -o pipefail; {joined}'"


def subprocess_with_output(cmd: List[str], cwd: Optional[str] = None,
                           env: Optional[dict] = None,
                           timeout: Optional[float] = None):
    """Run and capture (rc, stdout, stderr)."""
    proc = subprocess.run(cmd, cwd=cwd, env=env, capture_output=True,
                          text=True, timeout=timeout)
    return proc.returncode, proc.stdout, proc.stderr


def is_none_or_empty(value) -> bool:
    return value is None or (hasattr(value, "__len__") and len(value) == 0)


def is_not_empty(value) -> bool:
    return not is_none_or_empty(value)


def expand_env(value: str, env: Optional[dict] = None) -> str:
    """$VAR / ${VAR} expansion against os.environ (+ overrides)."""
    merged = dict(os.environ)
    if env:
        merged.update(env)
    return re.sub(
        r"\$\{(\w+)\}|\$(\w+)",
        lambda m: merged.get(m.group(1) or m.group(2),
                             m.group(0)),
        value)
