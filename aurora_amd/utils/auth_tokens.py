"""HMAC-signed identity tokens for REST and WebSocket auth.

The round-1 surface trusted a bare ``X-User-Id`` header and an unsigned
``<user_id>.<nonce>`` WS token — identity assertion, not authentication.
This module gives both surfaces real, verifiable identity the way the
reference does (signed JWT + Redis jti replay check,
reference: main_chatbot.py:107-158; identity-header anti-spoofing,
reference: tests/auth/test_identity_header_spoofing.py), using the shared
``INTERNAL_API_SECRET`` as the signing key instead of a JWT library.

Token wire format (one line, URL-safe):

    v1.<uid_b64>.<exp>.<jti>.<sig>

where ``uid_b64`` is base64url(user_id) without padding, ``exp`` is a unix
timestamp, ``jti`` a random 128-bit hex nonce (the replay-cache key), and
``sig`` = HMAC-SHA256(secret, "v1.<uid_b64>.<exp>.<jti>") hex digest.
Verification is constant-time and fails closed on any malformed field.
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import secrets as _secrets
import time
from typing import Optional


def _b64e(s: str) -> str:
    return base64.urlsafe_b64encode(s.encode()).decode().rstrip("=")


def _b64d(s: str) -> Optional[str]:
    try:
        pad = "=" * (-len(s) % 4)
        return base64.urlsafe_b64decode(s + pad).decode()
    except Exception:  # noqa: BLE001 - any malformed field fails closed
        return None


def _sig(secret: str, payload: str) -> str:
    return hmac.new(secret.encode(), payload.encode(), hashlib.sha256).hexdigest()


def mint_token(user_id: str, secret: str, ttl_s: int = 600,
               now: Optional[float] = None) -> str:
    """Sign a short-lived identity token for ``user_id``."""
    if not secret:
        raise ValueError("cannot mint tokens without a signing secret")
    exp = int((now if now is not None else time.time()) + ttl_s)
    jti = _secrets.token_hex(16)
    payload = f"v1.{_b64e(user_id)}.{exp}.{jti}"
    return f"{payload}.{_sig(secret, payload)}"


def verify_token(token: str, secret: str,
                 now: Optional[float] = None) -> Optional[dict]:
    """Returns ``{"user_id", "jti", "exp"}`` for a valid, unexpired token,
    else None. Constant-time signature compare; never raises."""
    if not secret or not token:
        return None
    parts = token.split(".")
    if len(parts) != 5 or parts[0] != "v1":
        return None
    _, uid_b64, exp_s, jti, sig = parts
    payload = f"v1.{uid_b64}.{exp_s}.{jti}"
    if not hmac.compare_digest(_sig(secret, payload), sig):
        return None
    try:
        exp = int(exp_s)
    except ValueError:
        return None
    if (now if now is not None else time.time()) >= exp:
        return None
    uid = _b64d(uid_b64)
    if not uid:
        return None
    return {"user_id": uid, "jti": jti, "exp": exp}


class ReplayCache:
    """In-memory jti one-shot cache (the reference's Redis jti SETNX,
    reference: main_chatbot.py:145). ``check_and_store`` returns False when
    the jti was already seen inside the window."""

    def __init__(self, window_s: float = 600.0):
        self.window_s = window_s
        self._seen: dict[str, float] = {}

    def check_and_store(self, jti: str, now: Optional[float] = None) -> bool:
        t = now if now is not None else time.time()
        if len(self._seen) > 4096:
            self._seen = {k: v for k, v in self._seen.items()
                          if t - v < self.window_s}
        if jti in self._seen and t - self._seen[jti] < self.window_s:
            return False
        self._seen[jti] = t
        return True
