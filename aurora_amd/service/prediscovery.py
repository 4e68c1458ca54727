"""Prediscovery + discovery: map infrastructure before incidents happen.

Rebuilds the reference's hourly prediscovery beat (reference:
chat/background/prediscovery_task.py:176-300 — a headless agent pre-maps
the environment and the result chunks are injected into RCA prompts) and
the discovery pipeline's graph writer (reference: services/discovery/ —
provider asset discovery + dependency inference -> Memgraph).

Without cloud credentials the discovery source is the local environment
(processes, listening sockets, k8s if kubectl answers) — the provider
modules plug in through the same `discover_*` interface.
"""
from __future__ import annotations

import asyncio
import json
import re
import time
from typing import Any, Dict, List


async def _sh(cmd: str, timeout: float = 20.0) -> str:
    try:
        proc = await asyncio.create_subprocess_shell(
            cmd, stdout=asyncio.subprocess.PIPE, stderr=asyncio.subprocess.DEVNULL)
        out, _ = await asyncio.wait_for(proc.communicate(), timeout=timeout)
        return out.decode("utf-8", "replace")
    except Exception:  # noqa: BLE001
        return ""


async def discover_local(graph) -> List[str]:
    """Local-host discovery: listening services become graph nodes."""
    found: List[str] = []
    out = await _sh("ss -ltnp 2>/dev/null || netstat -ltnp 2>/dev/null")
    for line in out.splitlines():
        m = re.search(r':(\d+)\s.*?users:\(\("([^"]+)"', line)
        if m:
            name = f"{m.group(2)}:{m.group(1)}"
            graph.upsert_service(name, kind="local_service", port=int(m.group(1)))
            found.append(name)
    return found


async def discover_kubernetes(graph) -> List[str]:
    found: List[str] = []
    out = await _sh("kubectl get svc -A -o json 2>/dev/null", timeout=15)
    if not out:
        return found
    try:
        doc = json.loads(out)
    except json.JSONDecodeError:
        return found
    for item in doc.get("items", []):
        meta = item.get("metadata", {})
        name = f"{meta.get('namespace','default')}/{meta.get('name','')}"
        graph.upsert_service(name, kind="k8s_service")
        found.append(name)
    return found


async def run_prediscovery(rt, args: Dict[str, Any]) -> dict:
    """Beat task: refresh the infra graph + store a prediscovery context
    artifact that build_rca_prompt can inject."""
    orgs = [args["org_id"]] if args.get("org_id") else [o["id"] for o in rt.admin.orgs()]
    total = 0
    for org_id in orgs:
        graph = rt.graph(org_id)
        store = rt.org(org_id)
        found = await discover_local(graph)
        found += await discover_kubernetes(graph)
        # dependency inference over the discovered assets (reference:
        # services/discovery/inference/ modules)
        from .discovery import discover_cloud_assets, run_inference
        inferred = await run_inference(graph)
        # provider asset discovery through the first org member with
        # connected cloud credentials (reference: providers/ discovery)
        assets = 0
        for u in rt.db.query("SELECT id FROM users WHERE org_id=?", [org_id]):
            if any(p.split(":")[0] in ("aws", "gcp", "azure")
                   for p in store.connected_providers(u["id"])):
                tctx = rt.tool_ctx(org_id, user_id=u["id"])
                assets = await discover_cloud_assets(tctx, graph)
                break
        ctx = {
            "discovered_at": time.time(),
            "services": [s["name"] for s in graph.list_services()][:100],
            "spof": graph.single_points_of_failure(),
            "stale": graph.stale_services()[:20],
            "inferred_edges": inferred,
            "provider_assets": assets,
        }
        store.save_artifact("prediscovery_context", json.dumps(ctx, indent=1))
        total += len(found)
    return {"discovered": total, "orgs": len(orgs)}


def prediscovery_context(store) -> str:
    row = store.get_artifact("prediscovery_context")
    return row["content"] if row else ""
