"""Secrets store — the KeyVault analogue.

The reference dereferences any ``*_keyvault_secret_id`` config value
from Azure KeyVault (reference convoy/keyvault.py:196
`parse_secret_ids`, 71 `fetch_credentials_conf`).  Locally, secrets
live in an encrypted JSON file; any config value ending in
``_secret_id`` is replaced by the named secret at load time.

Encryption: PBKDF2-HMAC-SHA256 key derivation + SHA256-counter-mode
keystream + HMAC-SHA256 authentication — stdlib-only (no cryptography
wheel in this image), authenticated-then-decrypted.  This protects
at-rest credentials on shared nodes; it is not a KMS.
"""
from __future__ import annotations

import hashlib
import hmac
import json
import os
from pathlib import Path
from typing import Any, Dict, Optional

_MAGIC = b"SYSEC1"
_ITER = 200_000


def _derive(passphrase: str, salt: bytes) -> bytes:
    return hashlib.pbkdf2_hmac("sha256", passphrase.encode(), salt, _ITER,
                               dklen=64)


def _keystream(key: bytes, n: int) -> bytes:
    out = bytearray()
    counter = 0
    while len(out) < n:
        out += hashlib.sha256(key + counter.to_bytes(8, "big")).digest()
        counter += 1
    return bytes(out[:n])


def _xor(data: bytes, ks: bytes) -> bytes:
    return bytes(a ^ b for a, b in zip(data, ks))


class SecretsStore:
    def __init__(self, path, passphrase: Optional[str] = None,
                 passphrase_env: str = "SHIPYARD_SECRETS_PASSPHRASE"):
        self.path = Path(path)
        self.passphrase = passphrase or os.environ.get(passphrase_env)
        if not self.passphrase:
            raise ValueError(
                f"secrets passphrase not provided (env {passphrase_env})")

    def _load(self) -> Dict[str, str]:
        if not self.path.exists():
            return {}
        raw = self.path.read_bytes()
        if raw[:6] != _MAGIC:
            raise ValueError("not a shipyard secrets file")
        salt, mac, ct = raw[6:22], raw[22:54], raw[54:]
        dk = _derive(self.passphrase, salt)
        enc_key, mac_key = dk[:32], dk[32:]
        want = hmac.new(mac_key, ct, hashlib.sha256).digest()
        if not hmac.compare_digest(mac, want):
            raise ValueError("secrets file MAC mismatch (bad passphrase "
                             "or corrupted file)")
        pt = _xor(ct, _keystream(enc_key, len(ct)))
        return json.loads(pt.decode())

    def _save(self, secrets: Dict[str, str]) -> None:
        salt = os.urandom(16)
        dk = _derive(self.passphrase, salt)
        enc_key, mac_key = dk[:32], dk[32:]
        pt = json.dumps(secrets).encode()
        ct = _xor(pt, _keystream(enc_key, len(pt)))
        mac = hmac.new(mac_key, ct, hashlib.sha256).digest()
        self.path.parent.mkdir(parents=True, exist_ok=True)
        self.path.write_bytes(_MAGIC + salt + mac + ct)
        self.path.chmod(0o600)

    def set(self, name: str, value: str) -> None:
        s = self._load()
        s[name] = value
        self._save(s)

    def get(self, name: str) -> str:
        s = self._load()
        if name not in s:
            raise KeyError(f"secret not found: {name}")
        return s[name]

    def delete(self, name: str) -> bool:
        s = self._load()
        if name in s:
            del s[name]
            self._save(s)
            return True
        return False

    def list(self):
        return sorted(self._load())


def parse_secret_ids(doc: Any, store: SecretsStore) -> Any:
    """Walk a config dict replacing any key ending `_secret_id` with the
    dereferenced plain key (reference convoy/keyvault.py:196)."""
    if isinstance(doc, dict):
        out = {}
        for k, v in doc.items():
            if isinstance(k, str) and k.endswith("_secret_id") \
                    and isinstance(v, str):
                out[k[:-len("_secret_id")]] = store.get(v)
            else:
                out[k] = parse_secret_ids(v, store)
        return out
    if isinstance(doc, list):
        return [parse_secret_ids(v, store) for v in doc]
    return doc
