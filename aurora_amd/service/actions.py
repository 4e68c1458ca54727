"""Actions: user-defined + system automations triggered on_incident or on a
schedule (reference: services/actions/executor.py:16,111; scheduler.py:38;
system actions postmortem_action.py / hpa_vpa right-sizing)."""
from __future__ import annotations

import json
import time
from typing import Dict


async def run_incident_actions(rt, org_id: str, incident_id: str) -> dict:
    store = rt.org(org_id)
    ran = []
    for action in store._select("actions", "trigger='on_incident' AND enabled=1"):
        try:
            await _execute(rt, org_id, action, incident_id=incident_id)
            ran.append(action["name"])
        except Exception as e:  # noqa: BLE001
            store.audit("action_error", f"{action['name']}: {e}")
    return {"ran": ran}


async def run_scheduled_actions(rt) -> dict:
    now = time.time()
    ran = []
    for org in rt.admin.orgs():
        store = rt.org(org["id"])
        for action in store._select("actions", "trigger='schedule' AND enabled=1"):
            interval = float(json.loads(action.get("config") or "{}")
                             .get("interval_s", 3600))
            if (action.get("last_run") or 0) + interval > now:
                continue
            try:
                await _execute(rt, org["id"], action)
                ran.append(action["name"])
            except Exception as e:  # noqa: BLE001
                store.audit("action_error", f"{action['name']}: {e}")
            store._update("actions", {"last_run": now}, "id=?", [action["id"]])
    return {"ran": ran}


async def _execute(rt, org_id: str, action: dict, incident_id: str = "") -> None:
    """System action kinds; 'notify' and 'agent_prompt' cover the custom
    automations, 'postmortem' is the auto-postmortem system action."""
    cfg = json.loads(action.get("config") or "{}")
    kind = cfg.get("kind", "notify")
    store = rt.org(org_id)
    if kind == "notify":
        from .notifications import dispatch_notification
        await dispatch_notification(rt, org_id, cfg.get("message", action["name"]),
                                    incident_id=incident_id)
    elif kind == "postmortem" and incident_id:
        inc = store.get_incident(incident_id)
        if inc and inc.get("summary"):
            from .postrca import build_postmortem
            pm = build_postmortem(inc, inc["summary"], store.citations(incident_id),
                                  [], store.lifecycle_events(incident_id))
            store.save_postmortem(incident_id, pm)
    elif kind == "alert_gap":
        # alert-coverage gap detection (reference: system alert-gap action):
        # services in the dependency graph that have NEVER produced an
        # alert-sourced incident have no monitoring wired to Aurora
        graph = rt.graph(org_id)
        covered = set()
        for inc in store.list_incidents(limit=500):
            svc = (json.loads(inc.get("payload") or "{}")
                   .get("service") or "")
            if isinstance(svc, dict):
                svc = svc.get("summary", "")
            if svc:
                covered.add(str(svc).split(".")[0].lower())
        gaps = [s["name"] for s in graph.list_services()
                if s["name"].split(":")[0].lower() not in covered]
        store.save_artifact("alert_gap_report", json.dumps({
            "generated_at": time.time(), "covered": sorted(covered),
            "uncovered_services": gaps[:100],
        }, indent=1))
        if gaps:
            from .notifications import dispatch_notification
            await dispatch_notification(
                rt, org_id,
                f"alert-coverage gap: {len(gaps)} graph services have no "
                f"alert source wired (see alert_gap_report artifact)",
                incident_id=incident_id)
    elif kind == "rightsizing":
        # scheduled right-sizing sweep: run the deterministic analysis over
        # kubectl top output when a cluster answers, persist the report
        # (reference: hpa_vpa scheduled action)
        from ..agent.tools.core import rightsizing_analysis
        from ..agent.tools.registry import ToolContext
        usage = ""
        try:
            import asyncio as _aio
            proc = await _aio.create_subprocess_shell(
                "kubectl top pods -A --no-headers 2>/dev/null",
                stdout=_aio.subprocess.PIPE)
            out, _ = await _aio.wait_for(proc.communicate(), timeout=15)
            usage = out.decode("utf-8", "replace")
        except Exception:  # noqa: BLE001
            usage = ""
        if usage.strip():
            report = await rightsizing_analysis(ToolContext(), usage)
            store.save_artifact("rightsizing_report", report)
        else:
            store.save_artifact("rightsizing_report",
                                "no cluster metrics available "
                                "(kubectl top unreachable)")
    elif kind == "agent_prompt":
        # headless one-shot agent run with a fixed prompt (right-sizing /
        # alert-gap style automations)
        rt.queue.enqueue("run_background_chat", {
            "org_id": org_id,
            "incident_id": incident_id or store.create_incident(
                f"action: {action['name']}", "action", {}),
            "prompt": cfg.get("prompt", ""), "max_turns": cfg.get("max_turns", 6)})
    store.lifecycle(incident_id or "-", f"action:{action['name']}", {"kind": kind})
