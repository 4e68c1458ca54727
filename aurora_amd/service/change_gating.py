"""Change gating: PR-risk review via a headless agent run.

Rebuilds the reference's GitHub/Bitbucket webhook -> investigate_pr flow
(reference: tasks/change_gating.py:293-477; dispatchers
tasks/github_webhook_tasks.py:1090, bitbucket_webhook_tasks.py:382): a PR
webhook enqueues a risk review; the agent assesses blast radius against
the infra graph and recent incidents; the verdict is persisted (and would
be posted back as a PR comment when the provider API is reachable).
"""
from __future__ import annotations

import json
from typing import Any, Dict, Optional

from ..engine.grammar import JsonGrammar
from ..engine.sampling_params import SamplingParams

REVIEW_PROMPT = """You are a change-risk reviewer for production deploys.
Given this pull request and operational context, reply ONLY a JSON object:
{{"verdict": "approve" | "caution" | "block", "risk_score": 0-100,
"reasons": ["..."]}}
PR: {pr}
Recently incident-linked services: {hot_services}
Services this repo maps to: {services}
JSON:"""


def normalize_pr_event(provider: str, payload: dict) -> Optional[dict]:
    """github / bitbucket payloads -> {repo, title, author, branch, files}."""
    if provider == "github":
        pr = payload.get("pull_request") or {}
        if not pr:
            return None
        return {
            "provider": "github",
            "repo": (payload.get("repository") or {}).get("full_name", ""),
            "number": payload.get("number") or pr.get("number"),
            "title": pr.get("title", ""),
            "author": (pr.get("user") or {}).get("login", ""),
            "branch": (pr.get("head") or {}).get("ref", ""),
            "additions": pr.get("additions", 0),
            "deletions": pr.get("deletions", 0),
            "changed_files": pr.get("changed_files", 0),
        }
    if provider == "bitbucket":
        pr = payload.get("pullrequest") or {}
        if not pr:
            return None
        return {
            "provider": "bitbucket",
            "repo": (payload.get("repository") or {}).get("full_name", ""),
            "number": pr.get("id"),
            "title": pr.get("title", ""),
            "author": (pr.get("author") or {}).get("display_name", ""),
            "branch": ((pr.get("source") or {}).get("branch") or {}).get("name", ""),
        }
    return None


async def investigate_pr(rt, args: Dict[str, Any]) -> dict:
    org_id = args["org_id"]
    pr = args["pr"]
    store = rt.org(org_id)
    graph = rt.graph(org_id)
    # context: services recently linked to incidents (the hot set)
    recent = store.list_incidents(limit=20)
    hot = sorted({json.loads(i.get("payload") or "{}").get("service", "")
                  for i in recent} - {""})
    repo_tail = (pr.get("repo") or "").split("/")[-1]
    services = [s["name"] for s in graph.list_services()
                if repo_tail and repo_tail in s["name"]] or [repo_tail]
    verdict: dict = {}
    try:
        res = await rt.llm(org_id).invoke(
            REVIEW_PROMPT.format(pr=json.dumps(pr)[:2000],
                                 hot_services=hot[:10], services=services),
            SamplingParams(temperature=0.1, max_tokens=250,
                           grammar=JsonGrammar(soft_close_bytes=190)),
            purpose="rca")
        verdict = json.loads(res.text)
    except Exception:  # noqa: BLE001
        verdict = {}
    if verdict.get("verdict") not in ("approve", "caution", "block"):
        # deterministic fallback heuristic: big changes touching hot services
        risk = min(100, int(pr.get("changed_files", 0) or 0) * 3
                   + (40 if set(services) & set(hot) else 0))
        verdict = {"verdict": "block" if risk >= 70 else
                   "caution" if risk >= 30 else "approve",
                   "risk_score": risk,
                   "reasons": ["heuristic fallback (judge unavailable)"]}
    record = dict(pr=pr, verdict=verdict, services=services)
    store.save_artifact(f"pr-review:{pr.get('repo')}#{pr.get('number')}",
                        json.dumps(record, indent=1))
    store.audit("pr_review", json.dumps({"repo": pr.get("repo"),
                                         "number": pr.get("number"),
                                         "verdict": verdict.get("verdict")}))
    rt.events.publish(org_id, {"type": "pr_review", **record})
    return verdict


def process_pr_webhook(rt, org_id: str, provider: str, payload: dict) -> Optional[str]:
    pr = normalize_pr_event(provider, payload)
    if pr is None:
        return None
    return rt.queue.enqueue(
        "investigate_pr", {"org_id": org_id, "pr": pr},
        dedup_key=f"pr:{pr.get('repo')}:{pr.get('number')}:"
                  f"{payload.get('action', payload.get('event', ''))}")
