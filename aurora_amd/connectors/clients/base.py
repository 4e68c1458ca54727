"""Shared plumbing for provider API clients.

The reference's observability/comms tools (reference:
tools/datadog_tool.py, query_newrelic.py, splunk_tool.py, …) each carry
the same skeleton: credential lookup, relative-time parsing, request
construction, pagination, response parsing, result truncation. This
module centralises that skeleton; each provider module contributes
``build_*`` (pure request construction) and ``parse_*`` (pure response
parsing) functions so every piece is testable against recorded fixtures
without egress, and an async ``run`` that drives them through a
pluggable transport.
"""
from __future__ import annotations

import datetime as _dt
import json
import re
from dataclasses import dataclass, field
from typing import Any, Awaitable, Callable, Dict, List, Optional


@dataclass
class Request:
    method: str
    url: str
    params: Dict[str, Any] = field(default_factory=dict)
    headers: Dict[str, str] = field(default_factory=dict)
    json_body: Optional[dict] = None

    def redacted(self) -> dict:
        """Loggable form: auth headers masked."""
        h = {k: ("***" if k.lower() in ("authorization", "dd-api-key",
                                        "dd-application-key", "api-key",
                                        "x-api-key", "authorization-token")
                 else v) for k, v in self.headers.items()}
        return {"method": self.method, "url": self.url, "params": self.params,
                "headers": h, "json": self.json_body}


#: transport: async (Request) -> dict (parsed JSON body). Tests inject a
#: fixture transport; production uses http_transport below; air-gapped
#: deployments get NoEgressError -> the tool's explicit degradation text.
Transport = Callable[[Request], Awaitable[dict]]


class ProviderAPIError(RuntimeError):
    pass


class NoEgressError(ProviderAPIError):
    pass


def http_transport(timeout_s: float = 20.0) -> Transport:
    """Real HTTP transport (httpx). Network failures surface as
    ProviderAPIError so tools degrade explicitly, never silently."""
    async def _send(req: Request) -> dict:
        import httpx
        try:
            async with httpx.AsyncClient(timeout=timeout_s) as cli:
                r = await cli.request(req.method, req.url, params=req.params,
                                      headers=req.headers, json=req.json_body)
        except httpx.TransportError as e:
            raise NoEgressError(f"network unreachable: {e}") from e
        if r.status_code >= 400:
            raise ProviderAPIError(f"HTTP {r.status_code}: {r.text[:300]}")
        try:
            return r.json()
        except json.JSONDecodeError:
            return r.text  # e.g. jenkins consoleText endpoints
    return _send


# ------------------------------------------------------------------ time

_REL = re.compile(r"^(?:now)?\s*-\s*(\d+)\s*(s|m|h|d|w)$", re.IGNORECASE)
_UNIT_S = {"s": 1, "m": 60, "h": 3600, "d": 86400, "w": 604800}


def parse_time(spec: Optional[str], now: Optional[float] = None) -> float:
    """'now', 'now-15m', '-2h', unix seconds, or ISO8601 -> unix seconds.
    (reference: datadog_tool.py:70-110 _parse_relative_time family)."""
    base = now if now is not None else _dt.datetime.now(
        _dt.timezone.utc).timestamp()
    if spec is None or spec == "" or spec.strip().lower() == "now":
        return base
    s = spec.strip()
    m = _REL.match(s)
    if m:
        return base - int(m.group(1)) * _UNIT_S[m.group(2).lower()]
    try:
        v = float(s)
        if v > 10_000_000_000:  # ms epoch
            return v / 1000.0
        return v
    except ValueError:
        pass
    try:
        return _dt.datetime.fromisoformat(s.replace("Z", "+00:00")).timestamp()
    except ValueError:
        raise ProviderAPIError(f"unparseable time spec: {spec!r}") from None


def window(time_from: Optional[str], time_to: Optional[str],
           default_span_s: int = 3600,
           now: Optional[float] = None) -> tuple:
    """(from_s, to_s) unix seconds; defaults to the last default_span_s."""
    t_to = parse_time(time_to, now=now)
    t_from = (parse_time(time_from, now=now) if time_from
              else t_to - default_span_s)
    if t_from >= t_to:
        raise ProviderAPIError(
            f"empty time window: from {t_from} >= to {t_to}")
    return t_from, t_to


def iso(ts: float) -> str:
    return _dt.datetime.fromtimestamp(ts, _dt.timezone.utc).strftime(
        "%Y-%m-%dT%H:%M:%SZ")


# -------------------------------------------------------------- summaries

def percentile(sorted_vals: List[float], q: float) -> Optional[float]:
    if not sorted_vals:
        return None
    idx = q * (len(sorted_vals) - 1)
    lo = int(idx)
    hi = min(lo + 1, len(sorted_vals) - 1)
    frac = idx - lo
    return sorted_vals[lo] * (1 - frac) + sorted_vals[hi] * frac


def series_stats(points: List[float]) -> dict:
    """min/max/avg/p95/last for a metric series (reference:
    datadog_tool.py:291-345 _summarize_series)."""
    vals = [v for v in points if v is not None]
    if not vals:
        return {"points": 0}
    sv = sorted(vals)
    return {"points": len(vals),
            "min": round(sv[0], 4), "max": round(sv[-1], 4),
            "avg": round(sum(vals) / len(vals), 4),
            "p95": round(percentile(sv, 0.95), 4),
            "last": round(vals[-1], 4)}


def fail_closed(fn):
    """Parser guard: a hostile or half-changed provider response must
    degrade into an explicit ProviderAPIError (which tools render as an
    ERROR observation), never leak a TypeError/KeyError into the agent
    turn."""
    import functools

    @functools.wraps(fn)
    def wrapped(*a, **k):
        try:
            return fn(*a, **k)
        except ProviderAPIError:
            raise
        except (AttributeError, TypeError, KeyError, ValueError,
                IndexError) as e:
            raise ProviderAPIError(
                f"malformed provider response ({type(e).__name__}: "
                f"{str(e)[:120]})") from e
    return wrapped


def truncate_items(items: List[str], max_items: int = 25,
                   max_chars: int = 8000) -> tuple:
    """(kept_items, truncated?) under both count and char budgets
    (reference: datadog_tool.py:123-137 _truncate_results)."""
    out: List[str] = []
    total = 0
    for it in items[:max_items]:
        if total + len(it) > max_chars:
            return out, True
        out.append(it)
        total += len(it)
    return out, len(items) > len(out)
