"""Connector registry: the 25-provider catalog under the uniform contract
(reference: server/connectors/ + AGENTS.md:47-82 — token storage via the
secrets backend with only a reference in the DB, per-connector skill,
connection-gated tools, webhook sources flagged).

API clients need egress this environment doesn't have; the contract —
connect/token-store/gating/catalog — is fully functional and tested, and
the gated tools surface degrades with explicit errors (tools/core.py).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..utils.secrets import secrets_backend


@dataclass
class ConnectorSpec:
    name: str
    category: str  # alerting | cloud | observability | code | comms | docs | infra
    auth: str  # token | oauth | webhook_only
    is_alert_source: bool = False
    tools: List[str] = field(default_factory=list)
    skill: str = ""


_CATALOG: Dict[str, ConnectorSpec] = {}


def _reg(spec: ConnectorSpec) -> None:
    _CATALOG[spec.name] = spec


for name, cat, auth, alert, tools in [
    ("pagerduty", "alerting", "token", True, []),
    ("opsgenie", "alerting", "token", True, ["query_opsgenie"]),
    ("datadog", "observability", "token", True, ["query_datadog"]),
    ("grafana", "observability", "token", True, ["query_grafana"]),
    ("newrelic", "observability", "token", True, ["query_newrelic"]),
    ("sentry", "observability", "token", True, ["query_sentry"]),
    ("splunk", "observability", "token", True, ["search_splunk"]),
    ("dynatrace", "observability", "token", True, ["query_dynatrace"]),
    ("bigpanda", "alerting", "token", True, []),
    ("netdata", "observability", "token", True, ["netdata"]),
    ("coroot", "observability", "token", True, ["coroot"]),
    ("thousandeyes", "observability", "token", False, ["thousandeyes"]),
    ("cloudflare", "infra", "token", False, ["cloudflare"]),
    ("incidentio", "alerting", "token", True, ["list_incidentio_incidents"]),
    ("aws", "cloud", "token", True, ["query_cloudwatch"]),
    ("gcp", "cloud", "oauth", False, []),
    ("azure", "cloud", "oauth", False, []),
    ("ovh", "cloud", "token", False, []),
    ("scaleway", "cloud", "token", False, []),
    ("flyio", "cloud", "token", False, ["query_flyio_metrics"]),
    ("tailscale", "infra", "token", False, []),
    ("kubernetes", "infra", "token", False, []),
    ("github", "code", "oauth", False, ["github_rca", "github_fix"]),
    ("bitbucket", "code", "oauth", False, []),
    ("jenkins", "code", "token", True, ["jenkins_rca"]),
    ("spinnaker", "code", "token", True, ["spinnaker_rca"]),
    ("cloudbees", "code", "token", True, ["cloudbees_rca"]),
    ("slack", "comms", "oauth", False, ["query_slack"]),
    ("jira", "docs", "oauth", True, ["query_jira"]),
    ("notion", "docs", "oauth", False, ["notion_search"]),
    ("confluence", "docs", "oauth", False, ["confluence_search"]),
    ("sharepoint", "docs", "oauth", False, []),
]:
    _reg(ConnectorSpec(name, cat, auth, alert, tools))


def connector_catalog() -> List[dict]:
    return [dict(name=c.name, category=c.category, auth=c.auth,
                 is_alert_source=c.is_alert_source, tools=c.tools)
            for c in _CATALOG.values()]


def get_connector(name: str) -> Optional[ConnectorSpec]:
    return _CATALOG.get(name)


def connect_provider(rt, store, user_id: str, provider: str, body: dict) -> dict:
    """Store credentials via the secrets backend; DB keeps only the ref
    (reference: store_tokens_in_db / vault ref format)."""
    # "<provider>:<account>" stores an alternate credential set for the
    # same connector (multi-account orgs — reference: AWS fan-out)
    base = provider.split(":", 1)[0]
    spec = _CATALOG.get(base)
    if spec is None:
        return {"error": f"unknown provider '{provider}'"}
    creds = body.get("credentials") or {}
    if spec.auth != "webhook_only" and not creds:
        return {"error": "credentials required"}
    ref = f"aurora/users/{store.org_id}/{user_id}/{provider}"
    if creds:
        secrets_backend().put(ref, creds)
    store.store_tokens(user_id, provider, ref)
    store.audit("connector_connected", f"{provider} by {user_id}")
    return {"connected": True, "provider": provider, "ref": ref}


def get_credentials(store, user_id: str, provider: str) -> Optional[dict]:
    ref = store.get_token_ref(user_id, provider)
    if ref is None:
        return None
    return secrets_backend().get(ref)
