"""Pluggable secrets backend (reference: utils/secrets/ — Vault KV v2 /
AWS SM behind SECRETS_BACKEND, ref format 'vault:kv/data/aurora/users/{n}').

Backends here: 'file' (authenticated-encrypted JSON at rest) and 'env'.
The store keeps REFERENCES (provider+user), never raw tokens, same as the
reference ('never log tokens', AGENTS.md:91).

At-rest construction (no third-party crypto in the image, stdlib only):
encrypt-then-MAC from HMAC-SHA256 primitives —

* per-entry random 16-byte salt;
* enc_key = HMAC(master, b"enc"|salt), mac_key = HMAC(master, b"mac"|salt)
  (an HKDF-extract-style domain separation);
* keystream = HMAC(enc_key, counter_le64) blocks — HMAC as a PRF in counter
  mode, so the pad is unique per entry and per write;
* tag = HMAC(mac_key, ref|salt|ciphertext), verified constant-time before
  decryption, binding each ciphertext to its ref (no entry swapping).

This replaces round 1's deterministic XOR keystream, which reused the pad
across rewrites of the same ref and had no integrity.
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import json
import os
import threading
from typing import Dict, Optional


def _hmac(key: bytes, msg: bytes) -> bytes:
    return hmac.new(key, msg, hashlib.sha256).digest()


def _keystream(enc_key: bytes, n: int) -> bytes:
    out = bytearray()
    counter = 0
    while len(out) < n:
        out += _hmac(enc_key, counter.to_bytes(8, "little"))
        counter += 1
    return bytes(out[:n])


def _seal(master: bytes, ref: str, raw: bytes) -> str:
    salt = os.urandom(16)
    enc_key = _hmac(master, b"enc" + salt)
    mac_key = _hmac(master, b"mac" + salt)
    ct = bytes(a ^ b for a, b in zip(raw, _keystream(enc_key, len(raw))))
    tag = _hmac(mac_key, ref.encode() + salt + ct)
    return "v2:" + base64.b64encode(salt + ct + tag).decode()


def _open(master: bytes, ref: str, blob: str) -> Optional[bytes]:
    try:
        data = base64.b64decode(blob[3:])
    except Exception:  # noqa: BLE001
        return None
    if len(data) < 48:  # salt(16) + tag(32)
        return None
    salt, ct, tag = data[:16], data[16:-32], data[-32:]
    mac_key = _hmac(master, b"mac" + salt)
    if not hmac.compare_digest(_hmac(mac_key, ref.encode() + salt + ct), tag):
        return None  # tampered / wrong key / swapped entry
    enc_key = _hmac(master, b"enc" + salt)
    return bytes(a ^ b for a, b in zip(ct, _keystream(enc_key, len(ct))))


def _legacy_keystream(key: bytes, n: int) -> bytes:
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
        enc = _seal(self.key, ref, json.dumps(value).encode())
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
        if enc.startswith("v2:"):
            raw = _open(self.key, ref, enc)
            if raw is None:
                return None
            try:
                return json.loads(raw)
            except Exception:  # noqa: BLE001
                return None
        # legacy round-1 entries (unauthenticated XOR): readable for
        # migration; rewritten in v2 on the next put
        try:
            raw = base64.b64decode(enc)
            ks = _legacy_keystream(self.key + ref.encode(), len(raw))
            return json.loads(bytes(a ^ b for a, b in zip(raw, ks)))
        except Exception:  # noqa: BLE001 - malformed legacy blob fails closed
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
