"""Secret scanning + redaction of tool outputs before they reach the model
or the UI (reference: utils/security/output_redaction.py:90,199 — regex
patterns + Shannon entropy)."""
from __future__ import annotations

import math
import re
from typing import List, Tuple

_PATTERNS: List[Tuple[str, re.Pattern]] = [
    ("aws_access_key", re.compile(r"\b(AKIA|ASIA)[0-9A-Z]{16}\b")),
    ("aws_secret_key", re.compile(r"(?i)aws_secret_access_key[\"'\s:=]+([A-Za-z0-9/+=]{40})")),
    # redact to END marker or end-of-text: tool-output truncation can cut
    # the END line and must not leak the key body
    ("private_key", re.compile(
        r"-----BEGIN [A-Z ]*PRIVATE KEY-----"
        r"[\s\S]*?(?:-----END [A-Z ]*PRIVATE KEY-----|\Z)")),
    ("github_token", re.compile(r"\b(ghp|gho|ghu|ghs|ghr)_[A-Za-z0-9]{36,}\b")),
    ("slack_token", re.compile(r"\bxox[baprs]-[A-Za-z0-9-]{10,}\b")),
    ("gcp_api_key", re.compile(r"\bAIza[0-9A-Za-z_\-]{35}\b")),
    ("jwt", re.compile(r"\beyJ[A-Za-z0-9_\-]{10,}\.[A-Za-z0-9_\-]{10,}\.[A-Za-z0-9_\-]{10,}\b")),
    ("bearer", re.compile(r"(?i)\b(authorization|bearer)[\"'\s:=]+[A-Za-z0-9._\-]{20,}")),
    ("password_kv", re.compile(r"(?i)\b(password|passwd|secret|api[_-]?key|token)[\"']?\s*[:=]\s*[\"']?([^\s\"',;]{8,})")),
    ("pg_url", re.compile(r"postgres(ql)?://[^\s:@]+:[^\s@]+@")),
]

_ENTROPY_TOKEN = re.compile(r"\b[A-Za-z0-9/+_\-]{28,}\b")


def shannon_entropy(s: str) -> float:
    if not s:
        return 0.0
    freq = {}
    for c in s:
        freq[c] = freq.get(c, 0) + 1
    n = len(s)
    return -sum(f / n * math.log2(f / n) for f in freq.values())


def redact(text: str, entropy_threshold: float = 4.4) -> str:
    """Replace secret-looking spans with [REDACTED:<kind>]."""
    out = text
    for kind, pat in _PATTERNS:
        out = pat.sub(f"[REDACTED:{kind}]", out)

    def _maybe(m: re.Match) -> str:
        tok = m.group(0)
        # skip obvious non-secrets: paths, hex hashes of known artefacts
        if "/" in tok and tok.count("/") > 2:
            return tok
        if shannon_entropy(tok) >= entropy_threshold:
            return "[REDACTED:high_entropy]"
        return tok

    return _ENTROPY_TOKEN.sub(_maybe, out)


def contains_secret(text: str) -> bool:
    return any(p.search(text) for _, p in _PATTERNS)
