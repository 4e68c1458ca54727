"""Datadog API client: logs / metrics / monitors / events queries.

Request shapes follow Datadog's public v1/v2 APIs as the reference tool
drives them (reference: tools/datadog_tool.py:138-568 — logs search v2
with cursor pagination, metrics query v1 with interval clamping and
series summarisation, monitor search, event stream).
"""
from __future__ import annotations

import json
from typing import Dict, List, Optional

from .base import (fail_closed,
                   ProviderAPIError, Request, Transport, iso, series_stats,
                   truncate_items, window)

MAX_LOG_PAGES = 3
LOGS_PAGE_LIMIT = 50
#: Datadog caps points per metrics query; pick interval so the window fits
#: (reference: datadog_tool.py:208-235 _clamp_interval/_pick_interval)
MAX_POINTS = 300


def _auth(creds: Dict[str, str]) -> Dict[str, str]:
    if not creds.get("api_key") or not creds.get("app_key"):
        raise ProviderAPIError("datadog credentials need api_key + app_key")
    return {"DD-API-KEY": creds["api_key"],
            "DD-APPLICATION-KEY": creds["app_key"]}


def _site(creds: Dict[str, str]) -> str:
    return f"https://api.{creds.get('site', 'datadoghq.com')}"


def pick_interval(span_s: float, caller_interval: Optional[int] = None) -> int:
    if caller_interval and span_s / caller_interval <= MAX_POINTS:
        return int(caller_interval)
    iv = max(1, int(span_s / MAX_POINTS))
    # snap up to a familiar bucket
    for b in (1, 5, 10, 30, 60, 300, 600, 1800, 3600, 14400, 86400):
        if iv <= b:
            return b
    return iv


# ------------------------------------------------------------------- logs

def build_logs_request(creds: dict, query: str, time_from: Optional[str],
                       time_to: Optional[str], cursor: Optional[str] = None,
                       limit: int = LOGS_PAGE_LIMIT,
                       now: Optional[float] = None) -> Request:
    f, t = window(time_from, time_to, default_span_s=900, now=now)
    body = {"filter": {"query": query or "*", "from": iso(f), "to": iso(t)},
            "page": {"limit": limit}, "sort": "-timestamp"}
    if cursor:
        body["page"]["cursor"] = cursor
    return Request("POST", f"{_site(creds)}/api/v2/logs/events/search",
                   headers=_auth(creds), json_body=body)


@fail_closed
def parse_logs_response(doc: dict) -> tuple:
    """-> (lines, next_cursor)."""
    lines: List[str] = []
    for ev in doc.get("data", []):
        a = ev.get("attributes", {})
        svc = a.get("service", "")
        st = a.get("status", "")
        msg = (a.get("message") or "").strip().replace("\n", " ")[:400]
        lines.append(f"[{a.get('timestamp', '')}] {svc} {st}: {msg}")
    cursor = (doc.get("meta", {}).get("page", {}) or {}).get("after")
    return lines, cursor


async def query_logs(transport: Transport, creds: dict, query: str,
                     time_from: Optional[str] = None,
                     time_to: Optional[str] = None,
                     now: Optional[float] = None) -> str:
    lines: List[str] = []
    cursor = None
    for _ in range(MAX_LOG_PAGES):  # cursor pagination
        doc = await transport(build_logs_request(
            creds, query, time_from, time_to, cursor, now=now))
        page, cursor = parse_logs_response(doc)
        lines.extend(page)
        if not cursor or not page:
            break
    kept, truncated = truncate_items(lines, max_items=80)
    head = f"datadog logs for {query!r}: {len(lines)} events"
    if truncated or cursor:
        head += " (truncated)"
    return head + "\n" + "\n".join(kept) if kept else head + " — no matches"


# ---------------------------------------------------------------- metrics

def build_metrics_request(creds: dict, query: str, time_from: Optional[str],
                          time_to: Optional[str],
                          now: Optional[float] = None) -> Request:
    f, t = window(time_from, time_to, default_span_s=3600, now=now)
    return Request("GET", f"{_site(creds)}/api/v1/query",
                   headers=_auth(creds),
                   params={"query": query, "from": int(f), "to": int(t)})


@fail_closed
def parse_metrics_response(doc: dict) -> str:
    if doc.get("status") == "error":
        raise ProviderAPIError(doc.get("error", "datadog query error"))
    out = []
    for s in doc.get("series", [])[:20]:
        pts = [p[1] for p in s.get("pointlist", []) if p and len(p) == 2]
        st = series_stats(pts)
        scope = s.get("scope", "*")
        unit = ""
        units = s.get("unit") or []
        if units and isinstance(units[0], dict):
            unit = units[0].get("short_name") or units[0].get("name") or ""
        out.append(f"{s.get('metric', '?')}{{{scope}}}"
                   + (f" [{unit}]" if unit else "") + f": {json.dumps(st)}")
    n = len(doc.get("series", []))
    head = f"datadog metrics: {n} series"
    return head + ("\n" + "\n".join(out) if out else " — empty result")


async def query_metrics(transport: Transport, creds: dict, query: str,
                        time_from: Optional[str] = None,
                        time_to: Optional[str] = None,
                        now: Optional[float] = None) -> str:
    doc = await transport(build_metrics_request(creds, query, time_from,
                                                time_to, now=now))
    return parse_metrics_response(doc)


# --------------------------------------------------------------- monitors

def build_monitors_request(creds: dict, query: str) -> Request:
    return Request("GET", f"{_site(creds)}/api/v1/monitor/search",
                   headers=_auth(creds),
                   params={"query": query or "status:alert", "per_page": 30})


@fail_closed
def parse_monitors_response(doc: dict) -> str:
    rows = []
    for m in doc.get("monitors", [])[:30]:
        rows.append(f"#{m.get('id')} [{m.get('status', '?')}] "
                    f"{m.get('name', '')} (type={m.get('type', '')}, "
                    f"last_triggered={m.get('last_triggered_ts', '')})")
    head = f"datadog monitors: {doc.get('metadata', {}).get('total_count', len(rows))} total"
    return head + ("\n" + "\n".join(rows) if rows else " — none match")


# ----------------------------------------------------------------- events

def build_events_request(creds: dict, time_from: Optional[str],
                         time_to: Optional[str], query: str = "",
                         now: Optional[float] = None) -> Request:
    f, t = window(time_from, time_to, default_span_s=3600, now=now)
    params = {"start": int(f), "end": int(t)}
    if query:
        params["tags"] = query
    return Request("GET", f"{_site(creds)}/api/v1/events",
                   headers=_auth(creds), params=params)


@fail_closed
def parse_events_response(doc: dict) -> str:
    rows = []
    for e in doc.get("events", [])[:40]:
        rows.append(f"[{e.get('date_happened', '')}] "
                    f"{e.get('alert_type', 'info')}: "
                    f"{(e.get('title') or '')[:160]}")
    return (f"datadog events: {len(doc.get('events', []))}"
            + ("\n" + "\n".join(rows) if rows else " — none"))


# --------------------------------------------------------------- dispatch

async def query(transport: Transport, creds: dict, query_type: str,
                query_str: str, time_from: Optional[str] = None,
                time_to: Optional[str] = None,
                now: Optional[float] = None) -> str:
    qt = (query_type or "logs").lower()
    if qt == "logs":
        return await query_logs(transport, creds, query_str, time_from,
                                time_to, now=now)
    if qt in ("metrics", "metric"):
        return await query_metrics(transport, creds, query_str, time_from,
                                   time_to, now=now)
    if qt == "monitors":
        return parse_monitors_response(
            await transport(build_monitors_request(creds, query_str)))
    if qt == "events":
        return parse_events_response(await transport(
            build_events_request(creds, time_from, time_to, query_str,
                                 now=now)))
    raise ProviderAPIError(
        f"unknown datadog query_type {query_type!r}: logs|metrics|monitors|events")
