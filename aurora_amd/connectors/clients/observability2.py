"""Second observability cluster: coroot, ThousandEyes, netdata,
Cloudflare, Opsgenie, incident.io, Fly.io metrics.

Same contract as observability.py (reference tools: coroot 924 LoC,
thousandeyes 554, netdata, cloudflare 939, query_opsgenie,
list_incidentio_incidents/timeline, query_flyio_metrics — registered
reference: cloud_tools.py:2093-2330,2849): pure build_*/parse_* pieces
through the pluggable transport.
"""
from __future__ import annotations

import json
from typing import Optional

from .base import (fail_closed,
                   ProviderAPIError, Request, Transport, iso, series_stats,
                   truncate_items, window)

# ------------------------------------------------------------------ coroot


def build_coroot_overview(creds: dict) -> Request:
    if not creds.get("base_url"):
        raise ProviderAPIError("coroot credentials need base_url")
    h = {}
    if creds.get("api_key"):
        h["X-API-Key"] = creds["api_key"]
    pid = creds.get("project", "default")
    return Request("GET", creds["base_url"].rstrip("/")
                   + f"/api/project/{pid}/overview/applications", headers=h)


@fail_closed
def parse_coroot_overview(doc) -> str:
    apps = doc if isinstance(doc, list) else doc.get("applications", [])
    rows = []
    for a in apps[:30]:
        status = a.get("status", "?")
        name = (a.get("id") or {}).get("name") if isinstance(
            a.get("id"), dict) else a.get("id", "?")
        indicators = ",".join(i.get("message", "")
                              for i in a.get("indicators", [])[:2])
        rows.append(f"[{status}] {name}" + (f": {indicators}"
                                            if indicators else ""))
    bad = sum(1 for a in apps if a.get("status") not in ("ok", "unknown"))
    return (f"coroot: {len(apps)} applications, {bad} unhealthy"
            + ("\n" + "\n".join(rows) if rows else ""))


async def coroot_query(transport: Transport, creds: dict, query: str,
                       **_kw) -> str:
    return parse_coroot_overview(await transport(build_coroot_overview(creds)))


# ------------------------------------------------------------ thousandeyes


def build_te_tests(creds: dict) -> Request:
    if not creds.get("token"):
        raise ProviderAPIError("thousandeyes credentials need token")
    return Request("GET", "https://api.thousandeyes.com/v7/tests",
                   headers={"Authorization": f"Bearer {creds['token']}"})


def build_te_results(creds: dict, test_id: str) -> Request:
    return Request("GET",
                   f"https://api.thousandeyes.com/v7/test-results/{test_id}/network",
                   headers={"Authorization": f"Bearer {creds['token']}"})


@fail_closed
def parse_te_results(doc: dict) -> str:
    results = doc.get("results", [])
    rows = []
    for r in results[:20]:
        rows.append(f"agent {r.get('agent', {}).get('agentName', '?')}: "
                    f"loss {r.get('loss', '?')}% "
                    f"latency {r.get('avgLatency', '?')}ms "
                    f"jitter {r.get('jitter', '?')}ms")
    loss = [float(r.get("loss", 0) or 0) for r in results]
    head = (f"thousandeyes network results: {len(results)} agents, "
            f"max loss {max(loss) if loss else 0}%")
    return head + ("\n" + "\n".join(rows) if rows else "")


async def thousandeyes_query(transport: Transport, creds: dict, query: str,
                             **_kw) -> str:
    q = (query or "").strip()
    if q and q != "tests":
        return parse_te_results(await transport(build_te_results(creds, q)))
    doc = await transport(build_te_tests(creds))
    rows = [f"{t.get('testId')}: {t.get('testName', '?')} "
            f"({t.get('type', '?')}, interval {t.get('interval', '?')}s)"
            for t in doc.get("tests", [])[:30]]
    return "thousandeyes tests:\n" + "\n".join(rows) if rows else \
        "thousandeyes: no tests configured"


# ---------------------------------------------------------------- netdata


def build_netdata_data(creds: dict, chart: str, time_from: Optional[str],
                       time_to: Optional[str],
                       now: Optional[float] = None) -> Request:
    if not creds.get("base_url"):
        raise ProviderAPIError("netdata credentials need base_url")
    f, t = window(time_from, time_to, default_span_s=3600, now=now)
    return Request("GET", creds["base_url"].rstrip("/") + "/api/v1/data",
                   params={"chart": chart or "system.cpu",
                           "after": int(f), "before": int(t),
                           "points": 120, "format": "json"})


@fail_closed
def parse_netdata_data(doc: dict) -> str:
    labels = doc.get("labels", [])
    data = doc.get("data", [])
    if not data:
        return "netdata: empty result"
    rows = []
    for di, name in enumerate(labels[1:], start=1):
        vals = [r[di] for r in data if r[di] is not None]
        rows.append(f"{name}: {json.dumps(series_stats(vals))}")
    return f"netdata {doc.get('id', '?')}:\n" + "\n".join(rows[:12])


async def netdata_query(transport: Transport, creds: dict, query: str,
                        time_from: Optional[str] = None,
                        time_to: Optional[str] = None,
                        now: Optional[float] = None) -> str:
    return parse_netdata_data(await transport(
        build_netdata_data(creds, query, time_from, time_to, now=now)))


# -------------------------------------------------------------- cloudflare


def build_cf_analytics(creds: dict, zone: str, time_from: Optional[str],
                       time_to: Optional[str],
                       now: Optional[float] = None) -> Request:
    if not creds.get("token"):
        raise ProviderAPIError("cloudflare credentials need token")
    f, t = window(time_from, time_to, default_span_s=3600, now=now)
    gql = """query($zone: String!, $since: Time!, $until: Time!) {
  viewer { zones(filter: {zoneTag: $zone}) {
    httpRequests1mGroups(limit: 60, filter: {datetime_geq: $since,
                                             datetime_lt: $until}) {
      sum { requests cachedRequests threats
            responseStatusMap { edgeResponseStatus requests } }
    } } } }"""
    return Request("POST", "https://api.cloudflare.com/client/v4/graphql",
                   headers={"Authorization": f"Bearer {creds['token']}"},
                   json_body={"query": gql,
                              "variables": {"zone": zone, "since": iso(f),
                                            "until": iso(t)}})


@fail_closed
def parse_cf_analytics(doc: dict) -> str:
    if doc.get("errors"):
        raise ProviderAPIError(
            "; ".join(str(e.get("message", e)) for e in doc["errors"])[:300])
    zones = (((doc.get("data") or {}).get("viewer") or {}).get("zones")) or []
    if not zones:
        return "cloudflare: zone not found"
    groups = zones[0].get("httpRequests1mGroups", [])
    total = sum(g["sum"].get("requests", 0) for g in groups)
    cached = sum(g["sum"].get("cachedRequests", 0) for g in groups)
    threats = sum(g["sum"].get("threats", 0) for g in groups)
    status: dict = {}
    for g in groups:
        for sm in g["sum"].get("responseStatusMap", []):
            k = str(sm.get("edgeResponseStatus"))
            status[k] = status.get(k, 0) + sm.get("requests", 0)
    errs = sum(v for k, v in status.items() if k.startswith("5"))
    top = sorted(status.items(), key=lambda kv: -kv[1])[:6]
    return (f"cloudflare: {total} requests, {cached} cached, "
            f"{threats} threats, {errs} 5xx\n"
            + ", ".join(f"{k}={v}" for k, v in top))


async def cloudflare_query(transport: Transport, creds: dict, query: str,
                           time_from: Optional[str] = None,
                           time_to: Optional[str] = None,
                           now: Optional[float] = None) -> str:
    zone = (query or creds.get("zone", "")).strip()
    if not zone:
        raise ProviderAPIError("cloudflare query needs a zone tag")
    return parse_cf_analytics(await transport(
        build_cf_analytics(creds, zone, time_from, time_to, now=now)))


# ---------------------------------------------------------------- opsgenie


def build_opsgenie_alerts(creds: dict, query: str) -> Request:
    if not creds.get("api_key"):
        raise ProviderAPIError("opsgenie credentials need api_key")
    return Request("GET", "https://api.opsgenie.com/v2/alerts",
                   headers={"Authorization": f"GenieKey {creds['api_key']}"},
                   params={"query": query or "status:open", "limit": 25,
                           "sort": "createdAt", "order": "desc"})


@fail_closed
def parse_opsgenie_alerts(doc: dict) -> str:
    rows = [f"[{a.get('priority', '?')}/{a.get('status', '?')}] "
            f"{a.get('message', '')[:140]} "
            f"(ack={a.get('acknowledged')}, at {a.get('createdAt', '')})"
            for a in doc.get("data", [])[:25]]
    return ("opsgenie alerts:" + ("\n" + "\n".join(rows) if rows
                                  else " none match"))


async def opsgenie_query(transport: Transport, creds: dict, query: str,
                         **_kw) -> str:
    return parse_opsgenie_alerts(await transport(
        build_opsgenie_alerts(creds, query)))


# --------------------------------------------------------------- incidentio


def build_incidentio_list(creds: dict) -> Request:
    if not creds.get("api_key"):
        raise ProviderAPIError("incidentio credentials need api_key")
    return Request("GET", "https://api.incident.io/v2/incidents",
                   headers={"Authorization": f"Bearer {creds['api_key']}"},
                   params={"page_size": 25})


def build_incidentio_timeline(creds: dict, incident_id: str) -> Request:
    return Request("GET", "https://api.incident.io/v2/incident_updates",
                   headers={"Authorization": f"Bearer {creds['api_key']}"},
                   params={"incident_id": incident_id, "page_size": 50})


@fail_closed
def parse_incidentio_list(doc: dict) -> str:
    rows = []
    for i in doc.get("incidents", [])[:25]:
        sev = (i.get("severity") or {}).get("name", "?")
        status = (i.get("incident_status") or {}).get("name", "?")
        rows.append(f"{i.get('reference', i.get('id', '?'))} [{sev}/{status}] "
                    f"{i.get('name', '')[:120]}")
    return ("incident.io:" + ("\n" + "\n".join(rows) if rows else " none"))


@fail_closed
def parse_incidentio_timeline(doc: dict) -> str:
    rows = [f"[{u.get('created_at', '')}] "
            f"{(u.get('new_incident_status') or {}).get('name', '')}: "
            f"{(u.get('message') or '')[:160]}"
            for u in doc.get("incident_updates", [])[:50]]
    return ("incident.io timeline:" + ("\n" + "\n".join(rows)
                                       if rows else " empty"))


async def incidentio_query(transport: Transport, creds: dict, query: str,
                           **_kw) -> str:
    q = (query or "").strip()
    if q and q != "list":
        return parse_incidentio_timeline(await transport(
            build_incidentio_timeline(creds, q)))
    return parse_incidentio_list(await transport(build_incidentio_list(creds)))


# ------------------------------------------------------------------- flyio


def build_flyio_metrics(creds: dict, app: str, promql: str,
                        time_from: Optional[str], time_to: Optional[str],
                        now: Optional[float] = None) -> Request:
    if not creds.get("token"):
        raise ProviderAPIError("flyio credentials need token")
    f, t = window(time_from, time_to, default_span_s=3600, now=now)
    org = creds.get("org", "personal")
    return Request("GET",
                   f"https://api.fly.io/prometheus/{org}/api/v1/query_range",
                   headers={"Authorization": f"Bearer {creds['token']}"},
                   params={"query": promql or
                           f'sum(rate(fly_app_http_responses_count{{app="{app}"}}[1m])) by (status)',
                           "start": int(f), "end": int(t), "step": 60})


@fail_closed
def parse_flyio_metrics(doc: dict) -> str:
    if doc.get("status") != "success":
        raise ProviderAPIError(str(doc.get("error", "prometheus error"))[:200])
    rows = []
    for series in (doc.get("data") or {}).get("result", [])[:12]:
        vals = [float(v[1]) for v in series.get("values", [])
                if v[1] not in (None, "NaN")]
        rows.append(f"{json.dumps(series.get('metric', {}))}: "
                    f"{json.dumps(series_stats(vals))}")
    return "fly.io metrics:" + ("\n" + "\n".join(rows) if rows
                                else " empty result")


async def flyio_query(transport: Transport, creds: dict, query: str,
                      time_from: Optional[str] = None,
                      time_to: Optional[str] = None,
                      now: Optional[float] = None) -> str:
    app = creds.get("app", "")
    return parse_flyio_metrics(await transport(build_flyio_metrics(
        creds, app, query, time_from, time_to, now=now)))
