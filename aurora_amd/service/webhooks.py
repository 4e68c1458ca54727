"""Alert-source webhook processing: 15 sources, signature verification,
payload normalisation, correlation, delayed RCA dispatch.

Rebuilds the per-source processors of the reference (reference:
routes/<source>/tasks.py — pagerduty/opsgenie/grafana/datadog/netdata/
splunk/dynatrace/bigpanda/newrelic/sentry/jenkins/jira/spinnaker/incidentio
+ routes/aws/cloudwatch_tasks.py; PagerDuty v3 signature check
pagerduty_routes.py:27).  Each normaliser maps a provider payload to
(title, severity, service, fingerprint); the shared pipeline is
correlate -> incident upsert -> trigger_delayed_rca (countdown).
"""
from __future__ import annotations

import hashlib
import hmac
import json
from typing import Any, Callable, Dict, Optional, Tuple

from .correlation import AlertCorrelator

# source -> normalizer(payload) -> dict(title, severity, service, fingerprint, description)
NORMALIZERS: Dict[str, Callable[[dict], dict]] = {}


def normalizer(source: str):
    def deco(fn):
        NORMALIZERS[source] = fn
        return fn
    return deco


def _g(d: Any, *path, default=""):
    for p in path:
        if isinstance(d, dict) and p in d:
            d = d[p]
        elif isinstance(d, list) and isinstance(p, int) and len(d) > p:
            d = d[p]
        else:
            return default
    return d


@normalizer("pagerduty")
def _pagerduty(p: dict) -> dict:
    ev = _g(p, "event", default={}) or p
    data = _g(ev, "data", default={}) or {}
    return {
        "title": data.get("title") or _g(data, "summary") or "PagerDuty incident",
        "severity": (_g(data, "priority", "summary") or "high").lower(),
        "service": _g(data, "service", "summary"),
        "fingerprint": data.get("id", ""),
        "description": _g(data, "body", "details"),
    }


@normalizer("datadog")
def _datadog(p: dict) -> dict:
    return {
        "title": p.get("title") or p.get("event_title") or "Datadog monitor alert",
        "severity": (p.get("priority") or p.get("alert_type") or "medium").lower(),
        "service": _g(p, "tags", default="") and next(
            (t.split(":", 1)[1] for t in p.get("tags", "").split(",")
             if t.startswith("service:")), "") or p.get("service", ""),
        "fingerprint": str(p.get("alert_id") or p.get("id") or ""),
        "description": p.get("body") or p.get("event_msg", ""),
    }


@normalizer("grafana")
def _grafana(p: dict) -> dict:
    a0 = _g(p, "alerts", 0, default={})
    return {
        "title": p.get("title") or _g(a0, "labels", "alertname") or "Grafana alert",
        "severity": _g(a0, "labels", "severity", default="medium"),
        "service": _g(a0, "labels", "service"),
        "fingerprint": a0.get("fingerprint", ""),
        "description": _g(a0, "annotations", "description"),
    }


@normalizer("cloudwatch")
def _cloudwatch(p: dict) -> dict:
    msg = p.get("Message")
    if isinstance(msg, str):
        try:
            msg = json.loads(msg)
        except json.JSONDecodeError:
            msg = {"raw": msg}
    msg = msg or p
    return {
        "title": msg.get("AlarmName") or "CloudWatch alarm",
        "severity": "high" if msg.get("NewStateValue") == "ALARM" else "medium",
        "service": _g(msg, "Trigger", "Namespace"),
        "fingerprint": msg.get("AlarmArn", ""),
        "description": msg.get("NewStateReason", ""),
    }


@normalizer("opsgenie")
def _opsgenie(p: dict) -> dict:
    a = _g(p, "alert", default={}) or {}
    return {"title": a.get("message") or "Opsgenie alert",
            "severity": (a.get("priority") or "P3").lower().replace("p1", "critical")
            .replace("p2", "high").replace("p3", "medium").replace("p4", "low")
            .replace("p5", "low"),
            "service": _g(a, "details", "service"),
            "fingerprint": a.get("alertId", ""),
            "description": a.get("description", "")}


@normalizer("netdata")
def _netdata(p: dict) -> dict:
    return {"title": p.get("name") or "Netdata alarm",
            "severity": (p.get("status") or "warning").lower()
            .replace("critical", "critical").replace("warning", "medium"),
            "service": p.get("host", ""), "fingerprint": str(p.get("alarm_id", "")),
            "description": p.get("info", "")}


@normalizer("splunk")
def _splunk(p: dict) -> dict:
    return {"title": p.get("search_name") or "Splunk alert",
            "severity": "high", "service": _g(p, "result", "host"),
            "fingerprint": p.get("sid", ""),
            "description": json.dumps(p.get("result", {}))[:500]}


@normalizer("dynatrace")
def _dynatrace(p: dict) -> dict:
    return {"title": p.get("ProblemTitle") or "Dynatrace problem",
            "severity": (p.get("ProblemSeverity") or "medium").lower(),
            "service": _g(p, "ImpactedEntities", 0, "name"),
            "fingerprint": str(p.get("PID", "")),
            "description": p.get("ProblemDetailsText", "")}


@normalizer("bigpanda")
def _bigpanda(p: dict) -> dict:
    return {"title": p.get("description") or "BigPanda incident",
            "severity": (p.get("status") or "medium").lower(),
            "service": p.get("host", ""), "fingerprint": p.get("id", ""),
            "description": json.dumps(p.get("tags", {}))[:500]}


@normalizer("newrelic")
def _newrelic(p: dict) -> dict:
    return {"title": p.get("condition_name") or p.get("title") or "New Relic alert",
            "severity": (p.get("severity") or p.get("priority") or "high").lower(),
            "service": p.get("targets", [{}])[0].get("name", "")
            if isinstance(p.get("targets"), list) else "",
            "fingerprint": str(p.get("incident_id", "")),
            "description": p.get("details", "")}


@normalizer("sentry")
def _sentry(p: dict) -> dict:
    return {"title": _g(p, "data", "issue", "title") or p.get("message")
            or "Sentry issue",
            "severity": (_g(p, "data", "issue", "level", default="error"))
            .replace("error", "high").replace("fatal", "critical"),
            "service": _g(p, "data", "issue", "project", "slug"),
            "fingerprint": str(_g(p, "data", "issue", "id")),
            "description": _g(p, "data", "issue", "culprit")}


@normalizer("jenkins")
def _jenkins(p: dict) -> dict:
    return {"title": f"Jenkins build failed: {p.get('name', '?')} "
            f"#{_g(p, 'build', 'number')}",
            "severity": "medium", "service": p.get("name", ""),
            "fingerprint": f"{p.get('name')}-{_g(p, 'build', 'number')}",
            "description": _g(p, "build", "full_url")}


@normalizer("jira")
def _jira(p: dict) -> dict:
    return {"title": _g(p, "issue", "fields", "summary") or "Jira issue",
            "severity": "low", "service": _g(p, "issue", "fields", "project", "key"),
            "fingerprint": _g(p, "issue", "key"),
            "description": str(_g(p, "issue", "fields", "description"))[:500]}


@normalizer("spinnaker")
def _spinnaker(p: dict) -> dict:
    return {"title": f"Spinnaker pipeline {_g(p, 'content', 'execution', 'status')}: "
            f"{_g(p, 'content', 'execution', 'name')}",
            "severity": "medium",
            "service": _g(p, "content", "execution", "application"),
            "fingerprint": _g(p, "content", "execution", "id"),
            "description": ""}


@normalizer("incidentio")
def _incidentio(p: dict) -> dict:
    inc = _g(p, "incident", default={}) or {}
    return {"title": inc.get("name") or "incident.io incident",
            "severity": (_g(inc, "severity", "name") or "medium").lower(),
            "service": "", "fingerprint": inc.get("id", ""),
            "description": inc.get("summary", "")}


# ---- signature verification (per-provider HMAC) ---------------------------

def verify_signature(source: str, body: bytes, headers: Dict[str, str],
                     secret: Optional[str]) -> bool:
    """HMAC checks per source (reference: pagerduty_routes.py:27 v3 check).
    No secret configured => accept (the reference's default for most dev
    sources) but callers audit it."""
    if not secret:
        return True
    h = {k.lower(): v for k, v in headers.items()}
    if source == "pagerduty":
        sig = h.get("x-pagerduty-signature", "")
        want = "v1=" + hmac.new(secret.encode(), body, hashlib.sha256).hexdigest()
        return hmac.compare_digest(sig, want)
    if source in ("github", "bitbucket"):
        sig = h.get("x-hub-signature-256", "")
        want = "sha256=" + hmac.new(secret.encode(), body, hashlib.sha256).hexdigest()
        return hmac.compare_digest(sig, want)
    # generic: X-Signature: hex hmac-sha256
    sig = h.get("x-signature", "")
    want = hmac.new(secret.encode(), body, hashlib.sha256).hexdigest()
    return hmac.compare_digest(sig, want)


# ---- the shared pipeline --------------------------------------------------

def process_webhook(rt, org_id: str, source: str, payload: dict,
                    user_id: Optional[str] = None,
                    rca_countdown_s: float = 1.0,
                    rca_args: Optional[dict] = None) -> Tuple[str, bool]:
    """Normalise -> correlate -> upsert incident -> dispatch delayed RCA.
    Returns (incident_id, created_new). ``rca_args`` (e.g. max_turns,
    tool_subset) are forwarded to the background investigation."""
    norm_fn = NORMALIZERS.get(source)
    if norm_fn is None:
        raise ValueError(f"unknown alert source '{source}'")
    norm = norm_fn(payload or {})
    # defensive caps: a hostile/buggy alert source must not balloon
    # incident rows or the correlator's similarity text
    norm["title"] = str(norm.get("title") or "(untitled alert)")[:500]
    if norm.get("external_id") is not None:
        norm["external_id"] = str(norm["external_id"])[:200]
    store = rt.org(org_id)
    corr = rt.correlator(org_id)
    enriched = dict(payload or {}, **{k: v for k, v in norm.items() if v})
    iid, created = corr.handle_alert(source, norm["title"], enriched,
                                     fingerprint=norm.get("fingerprint", ""),
                                     user_id=user_id)
    if created:
        # delayed RCA: the countdown lets provider runbook fields settle
        # (reference: trigger_delayed_rca countdown, tasks.py:235)
        rt.queue.enqueue("trigger_delayed_rca",
                         dict(rca_args or {}, org_id=org_id, incident_id=iid,
                              user_id=user_id),
                         countdown_s=rca_countdown_s,
                         dedup_key=f"delayed-rca:{iid}")
        rt.events.publish(org_id, {"type": "incident_created", "incident_id": iid,
                                   "title": norm["title"], "source": source})
    return iid, created


def sources() -> list:
    return sorted(NORMALIZERS)
