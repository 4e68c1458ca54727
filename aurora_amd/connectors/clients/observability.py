"""New Relic, Splunk, Sentry, Dynatrace, Grafana and CloudWatch clients.

Each provider contributes pure build_*/parse_* pieces plus an async
``query`` entry driven through the shared transport (reference tools:
query_newrelic.py — NerdGraph NRQL; splunk_tool.py — oneshot search
jobs + index listing; query_sentry.py — org issue search; dynatrace —
problems/metrics v2; cloudwatch — metric statistics / log insights).
"""
from __future__ import annotations

import json
from typing import Dict, List, Optional

from .base import (fail_closed,
                   ProviderAPIError, Request, Transport, iso, series_stats,
                   truncate_items, window)

# ------------------------------------------------------------- New Relic


def build_nrql_request(creds: dict, nrql: str) -> Request:
    if not creds.get("api_key"):
        raise ProviderAPIError("newrelic credentials need api_key")
    if not creds.get("account_id"):
        raise ProviderAPIError("newrelic credentials need account_id")
    gql = ("{ actor { account(id: %s) { nrql(query: %s) "
           "{ results } } } }" % (int(creds["account_id"]),
                                  json.dumps(nrql)))
    return Request("POST", creds.get("graphql_url",
                                     "https://api.newrelic.com/graphql"),
                   headers={"API-Key": creds["api_key"],
                            "Content-Type": "application/json"},
                   json_body={"query": gql})


@fail_closed
def parse_nrql_response(doc: dict) -> str:
    if doc.get("errors"):
        raise ProviderAPIError(
            "; ".join(e.get("message", "?") for e in doc["errors"])[:300])
    results = (((doc.get("data") or {}).get("actor") or {})
               .get("account") or {}).get("nrql", {}).get("results", [])
    rows = [json.dumps(r)[:300] for r in results[:40]]
    kept, trunc = truncate_items(rows)
    head = f"newrelic NRQL: {len(results)} rows" + (" (truncated)" if trunc else "")
    return head + ("\n" + "\n".join(kept) if kept else " — empty result")


async def newrelic_query(transport: Transport, creds: dict, nrql: str,
                         **_kw) -> str:
    return parse_nrql_response(await transport(build_nrql_request(creds, nrql)))


# ---------------------------------------------------------------- Splunk


def normalize_spl(query: str) -> str:
    """Splunk oneshot searches must start with a generating command
    (reference: splunk_tool.py prepends 'search ')."""
    q = (query or "").strip()
    if not q:
        raise ProviderAPIError("empty splunk search")
    if q.startswith("|") or q.lower().startswith("search "):
        return q
    return "search " + q


def build_splunk_search(creds: dict, query: str, time_from: Optional[str],
                        time_to: Optional[str], limit: int = 100,
                        now: Optional[float] = None) -> Request:
    if not creds.get("base_url") or not creds.get("token"):
        raise ProviderAPIError("splunk credentials need base_url + token")
    f, t = window(time_from, time_to, default_span_s=3600, now=now)
    return Request(
        "POST", creds["base_url"].rstrip("/") + "/services/search/jobs",
        headers={"Authorization": f"Bearer {creds['token']}"},
        params={"output_mode": "json"},
        json_body={"search": normalize_spl(query), "exec_mode": "oneshot",
                   "earliest_time": iso(f), "latest_time": iso(t),
                   "count": limit})


@fail_closed
def parse_splunk_results(doc: dict) -> str:
    rows = []
    for r in doc.get("results", [])[:60]:
        raw = r.get("_raw")
        if raw:
            rows.append(str(raw)[:300])
        else:
            rows.append(json.dumps(r)[:300])
    kept, trunc = truncate_items(rows, max_items=60)
    head = f"splunk: {len(doc.get('results', []))} results" + \
        (" (truncated)" if trunc else "")
    return head + ("\n" + "\n".join(kept) if kept else " — no events")


def build_splunk_indexes(creds: dict) -> Request:
    return Request("GET",
                   creds["base_url"].rstrip("/") + "/services/data/indexes",
                   headers={"Authorization": f"Bearer {creds['token']}"},
                   params={"output_mode": "json", "count": 50})


@fail_closed
def parse_splunk_indexes(doc: dict) -> str:
    names = [e.get("name", "?") for e in doc.get("entry", [])]
    return "splunk indexes: " + (", ".join(names) if names else "(none)")


async def splunk_query(transport: Transport, creds: dict, query: str,
                       time_from: Optional[str] = None,
                       time_to: Optional[str] = None,
                       now: Optional[float] = None) -> str:
    if (query or "").strip().lower() in ("indexes", "list indexes"):
        return parse_splunk_indexes(await transport(build_splunk_indexes(creds)))
    return parse_splunk_results(await transport(
        build_splunk_search(creds, query, time_from, time_to, now=now)))


# ---------------------------------------------------------------- Sentry


def build_sentry_issues(creds: dict, query: str,
                        stats_period: str = "24h") -> Request:
    if not creds.get("token") or not creds.get("org"):
        raise ProviderAPIError("sentry credentials need token + org")
    base = creds.get("base_url", "https://sentry.io").rstrip("/")
    return Request(
        "GET", f"{base}/api/0/organizations/{creds['org']}/issues/",
        headers={"Authorization": f"Bearer {creds['token']}"},
        params={"query": query or "is:unresolved", "statsPeriod": stats_period,
                "limit": 25, "sort": "freq"})


@fail_closed
def parse_sentry_issues(doc) -> str:
    if isinstance(doc, dict):  # error envelope
        raise ProviderAPIError(str(doc.get("detail", doc))[:300])
    rows = []
    for i in doc[:25]:
        rows.append(f"{i.get('shortId', '?')} [{i.get('level', '?')}] "
                    f"x{i.get('count', '?')} {i.get('title', '')[:160]} "
                    f"(last: {i.get('lastSeen', '')})")
    return (f"sentry issues: {len(doc)}"
            + ("\n" + "\n".join(rows) if rows else " — none match"))


async def sentry_query(transport: Transport, creds: dict, query: str,
                       **_kw) -> str:
    return parse_sentry_issues(await transport(
        build_sentry_issues(creds, query)))


# ------------------------------------------------------------- Dynatrace


def build_dynatrace_problems(creds: dict, time_from: Optional[str],
                             time_to: Optional[str],
                             now: Optional[float] = None) -> Request:
    if not creds.get("base_url") or not creds.get("token"):
        raise ProviderAPIError("dynatrace credentials need base_url + token")
    f, t = window(time_from, time_to, default_span_s=7200, now=now)
    return Request("GET", creds["base_url"].rstrip("/") + "/api/v2/problems",
                   headers={"Authorization": f"Api-Token {creds['token']}"},
                   params={"from": int(f * 1000), "to": int(t * 1000),
                           "pageSize": 25})


@fail_closed
def parse_dynatrace_problems(doc: dict) -> str:
    rows = []
    for p in doc.get("problems", [])[:25]:
        impacted = ",".join(e.get("name", "?")
                            for e in p.get("impactedEntities", [])[:3])
        rows.append(f"{p.get('displayId', '?')} [{p.get('status', '?')}/"
                    f"{p.get('severityLevel', '?')}] {p.get('title', '')[:140]}"
                    + (f" impacting {impacted}" if impacted else ""))
    return (f"dynatrace problems: {doc.get('totalCount', len(rows))}"
            + ("\n" + "\n".join(rows) if rows else " — none in window"))


def build_dynatrace_metrics(creds: dict, selector: str,
                            time_from: Optional[str], time_to: Optional[str],
                            now: Optional[float] = None) -> Request:
    f, t = window(time_from, time_to, default_span_s=3600, now=now)
    return Request("GET",
                   creds["base_url"].rstrip("/") + "/api/v2/metrics/query",
                   headers={"Authorization": f"Api-Token {creds['token']}"},
                   params={"metricSelector": selector,
                           "from": int(f * 1000), "to": int(t * 1000),
                           "resolution": "120"})


@fail_closed
def parse_dynatrace_metrics(doc: dict) -> str:
    rows = []
    for res in doc.get("result", [])[:10]:
        mid = res.get("metricId", "?")
        for d in res.get("data", [])[:10]:
            st = series_stats([v for v in d.get("values", [])
                               if v is not None])
            dims = ",".join(map(str, d.get("dimensions", [])))
            rows.append(f"{mid}{{{dims}}}: {json.dumps(st)}")
    return ("dynatrace metrics:"
            + ("\n" + "\n".join(rows) if rows else " empty result"))


async def dynatrace_query(transport: Transport, creds: dict, query: str,
                          time_from: Optional[str] = None,
                          time_to: Optional[str] = None,
                          now: Optional[float] = None) -> str:
    q = (query or "problems").strip()
    if q == "problems" or q.startswith("problems"):
        return parse_dynatrace_problems(await transport(
            build_dynatrace_problems(creds, time_from, time_to, now=now)))
    return parse_dynatrace_metrics(await transport(
        build_dynatrace_metrics(creds, q, time_from, time_to, now=now)))


# --------------------------------------------------------------- Grafana


def build_grafana_search(creds: dict, query: str) -> Request:
    if not creds.get("base_url") or not creds.get("token"):
        raise ProviderAPIError("grafana credentials need base_url + token")
    return Request("GET", creds["base_url"].rstrip("/") + "/api/search",
                   headers={"Authorization": f"Bearer {creds['token']}"},
                   params={"query": query or "", "limit": 30})


@fail_closed
def parse_grafana_search(doc) -> str:
    rows = [f"[{d.get('type', '?')}] {d.get('title', '?')} "
            f"(uid={d.get('uid', '?')}, folder={d.get('folderTitle', '')})"
            for d in doc[:30]]
    return ("grafana dashboards:"
            + ("\n" + "\n".join(rows) if rows else " none match"))


async def grafana_query(transport: Transport, creds: dict, query: str,
                        **_kw) -> str:
    return parse_grafana_search(await transport(
        build_grafana_search(creds, query)))


# ------------------------------------------------------------ CloudWatch
# (SigV4 signing needs botocore; without it the tool degrades explicitly.
# The query construction/parsing is still exercised in tests via the
# transport seam.)


def build_cloudwatch_metric(creds: dict, namespace: str, metric: str,
                            dimensions: Dict[str, str],
                            time_from: Optional[str], time_to: Optional[str],
                            stat: str = "Average", period: int = 300,
                            now: Optional[float] = None) -> Request:
    region = creds.get("region", "us-east-1")
    f, t = window(time_from, time_to, default_span_s=3600, now=now)
    dims = [{"Name": k, "Value": v} for k, v in sorted(dimensions.items())]
    body = {"Namespace": namespace,
            "MetricName": metric,
            "Dimensions": dims,
            "StartTime": iso(f), "EndTime": iso(t),
            "Period": period, "Statistics": [stat]}
    return Request("POST", f"https://monitoring.{region}.amazonaws.com/",
                   headers={"X-Amz-Target":
                            "GraniteServiceVersion20100801.GetMetricStatistics",
                            "Content-Type": "application/x-amz-json-1.0"},
                   json_body=body)


@fail_closed
def parse_cloudwatch_metric(doc: dict) -> str:
    pts = sorted(doc.get("Datapoints", []),
                 key=lambda p: p.get("Timestamp", ""))
    vals = [p.get("Average", p.get("Sum", p.get("Maximum")))
            for p in pts]
    st = series_stats([v for v in vals if v is not None])
    return (f"cloudwatch {doc.get('Label', '?')}: {json.dumps(st)}")
