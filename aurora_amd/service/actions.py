"""Actions: user-defined + system automations triggered on_incident or on a
schedule (reference: services/actions/executor.py:16,111; scheduler.py:38;
system actions postmortem_action.py / hpa_vpa right-sizing)."""
from __future__ import annotations

import json
import time
from typing import Any, Dict


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
    elif kind == "agent_prompt":
        # headless one-shot agent run with a fixed prompt (right-sizing /
        # alert-gap style automations)
        rt.queue.enqueue("run_background_chat", {
            "org_id": org_id,
            "incident_id": incident_id or store.create_incident(
                f"action: {action['name']}", "action", {}),
            "prompt": cfg.get("prompt", ""), "max_turns": cfg.get("max_turns", 6)})
    store.lifecycle(incident_id or "-", f"action:{action['name']}", {"kind": kind})
