"""kubectl-agent cluster registry: token registration + resolution."""
from __future__ import annotations

from typing import Optional

from ..kubectl_agent.agent import cluster_id_from_token
from ..utils.secrets import secrets_backend


def register_kubectl_agent(rt, org_id: str, token: str) -> str:
    """Admin registers a cluster token; the agent later dials in with it."""
    cid = cluster_id_from_token(token)
    ref = f"aurora/kubectl_agent/{org_id}/{cid}"
    secrets_backend().put(ref, {"token": token})
    rt.org(org_id).store_tokens("system", f"kubectl_agent:{cid}", ref)
    return cid


def resolve_agent_token(rt, token: str) -> Optional[str]:
    """Token -> org id, verified against the stored secret."""
    cid = cluster_id_from_token(token)
    for org in rt.admin.orgs():
        store = rt.org(org["id"])
        ref = store.get_token_ref("system", f"kubectl_agent:{cid}")
        if ref:
            data = secrets_backend().get(ref)
            if data and data.get("token") == token:
                return org["id"]
    return None
