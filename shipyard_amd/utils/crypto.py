"""Crypto helpers (analogue of reference convoy/crypto.py).

SSH keypair generation (ssh-keygen subprocess, reference crypto.py:127
`generate_ssh_keypair`), remote command execution over ssh (crypto.py:
171 `connect_or_exec_ssh_command`), and string encryption for on-node
credentials — the RSA/PFX machinery of the reference maps to the
authenticated secrets-store cipher (shipyard_amd/config/secrets.py),
since there is no Azure certificate store to round-trip through.
"""
from __future__ import annotations

import shutil
import subprocess
from pathlib import Path
from typing import List, Optional, Tuple

from shipyard_amd import utils

logger = utils.get_logger(__name__)

DEFAULT_KEY_NAME = "id_rsa_shipyard"


def generate_ssh_keypair(export_path, prefix: str = DEFAULT_KEY_NAME,
                         comment: str = "shipyard-amd") -> Tuple[Path, Path]:
    """ssh-keygen an RSA keypair; returns (private, public) paths."""
    if shutil.which("ssh-keygen") is None:
        raise RuntimeError("ssh-keygen not installed")
    priv = Path(export_path) / prefix
    priv.parent.mkdir(parents=True, exist_ok=True)
    if priv.exists():
        priv.unlink()
    pub = Path(str(priv) + ".pub")
    if pub.exists():
        pub.unlink()
    subprocess.run(
        ["ssh-keygen", "-f", str(priv), "-t", "rsa", "-b", "3072", "-N", "",
         "-C", comment, "-q"], check=True)
    priv.chmod(0o600)
    return priv, pub


def ssh_command(host: str, command: str, username: Optional[str] = None,
                private_key: Optional[str] = None,
                extra_options: Optional[List[str]] = None) -> List[str]:
    """Synthesize an ssh exec command line (reference crypto.py:171)."""
    cmd = ["ssh", "-o", "StrictHostKeyChecking=no",
           "-o", "UserKnownHostsFile=/dev/null"]
    if private_key:
        cmd += ["-i", private_key]
    cmd += extra_options or []
    target = f"{username}@{host}" if username else host
    cmd += [target, command]
    return cmd
