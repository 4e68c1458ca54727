"""Pluggable secrets backend (reference: utils/secrets/ — Vault KV v2 /
AWS SM behind SECRETS_BACKEND, ref format 'vault:kv/data/aurora/users/{n}').

Backends here: 'file' (encrypted-at-rest JSON using an XOR keystream keyed
from AURORA_SECRET_KEY — no cryptography dep guaranteed in the image) and
'env'.  The store keeps REFERENCES (provider+user), never raw tokens, same
as the reference ('never log tokens', AGENTS.md:91)."""
from __future__ import annotations

import base64
import hashlib
import json
import os
import threading
from typing import Dict, Optional


def _keystream(key: bytes, n: int) -> bytes:
    out = b""
    counter = 0
    while len(out) < n:
        out += hashlib.sha256(key + counter.to_bytes(8, "little")).digest()
        counter += 1
    return out[:n]


class SecretsBackend:
    def put(self, ref: str, value: Dict) -> None:
        raise NotImplementedError

    def get(self, ref: str) -> Optional[Dict]:
        raise NotImplementedError


class FileSecretsBackend(SecretsBackend):
    def __init__(self, path: str, key: Optional[str] = None):
        self.path = path
        self.key = (key or os.environ.get("AURORA_SECRET_KEY", "dev-insecure-key")
                    ).encode()
        self._lock = threading.Lock()

    def _load(self) -> Dict[str, str]:
        if not os.path.exists(self.path):
            return {}
        with open(self.path) as f:
            return json.load(f)

    def put(self, ref: str, value: Dict) -> None:
        raw = json.dumps(value).encode()
        ks = _keystream(self.key + ref.encode(), len(raw))
        enc = base64.b64encode(bytes(a ^ b for a, b in zip(raw, ks))).decode()
        with self._lock:
            data = self._load()
            data[ref] = enc
            with open(self.path, "w") as f:
                json.dump(data, f)
            os.chmod(self.path, 0o600)

    def get(self, ref: str) -> Optional[Dict]:
        with self._lock:
            data = self._load()
        enc = data.get(ref)
        if enc is None:
            return None
        raw = base64.b64decode(enc)
        ks = _keystream(self.key + ref.encode(), len(raw))
        try:
            return json.loads(bytes(a ^ b for a, b in zip(raw, ks)))
        except Exception:
            return None


class EnvSecretsBackend(SecretsBackend):
    def put(self, ref: str, value: Dict) -> None:
        raise RuntimeError("env secrets backend is read-only")

    def get(self, ref: str) -> Optional[Dict]:
        v = os.environ.get("AURORA_SECRET_" + ref.replace("/", "_").replace(":", "_").upper())
        return json.loads(v) if v else None


_backend: Optional[SecretsBackend] = None


def secrets_backend() -> SecretsBackend:
    global _backend
    if _backend is None:
        from .config import settings
        kind = os.environ.get("SECRETS_BACKEND", "file")
        if kind == "env":
            _backend = EnvSecretsBackend()
        else:
            _backend = FileSecretsBackend(settings().secrets_path)
    return _backend


def reset_backend() -> None:
    global _backend
    _backend = None
