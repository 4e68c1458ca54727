"""Log sanitisation: never log tokens/secrets (reference:
utils/log_sanitizer.py, utils/auth/log_redact.py, AGENTS.md:91).  Installs
a logging.Filter that redacts via the guardrails secret scanner; stdout-only
handler for container log aggregation (reference: celery_config.py:8-24)."""
from __future__ import annotations

import logging
import sys

from ..guardrails.redaction import redact


class RedactingFilter(logging.Filter):
    def filter(self, record: logging.LogRecord) -> bool:
        try:
            msg = record.getMessage()
            clean = redact(msg)
            if clean != msg:
                record.msg = clean
                record.args = ()
        except Exception:  # noqa: BLE001 - never let logging break the app
            pass
        return True


def configure_logging(level: int = logging.INFO) -> None:
    root = logging.getLogger()
    root.setLevel(level)
    if not any(isinstance(h, logging.StreamHandler) for h in root.handlers):
        h = logging.StreamHandler(sys.stdout)  # stdout-only (INC-445 rationale)
        h.setFormatter(logging.Formatter(
            "%(asctime)s %(levelname)s %(name)s %(message)s"))
        root.addHandler(h)
    for h in root.handlers:
        if not any(isinstance(f, RedactingFilter) for f in h.filters):
            h.addFilter(RedactingFilter())
    # noisy-logger suppression (reference: main_chatbot.py:7-16)
    for noisy in ("httpx", "httpcore", "asyncio", "uvicorn.access"):
        logging.getLogger(noisy).setLevel(logging.WARNING)
