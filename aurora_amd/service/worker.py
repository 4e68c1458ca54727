"""Background investigation pipeline: the run_background_chat rebuild.

Task flow mirrors the reference exactly (reference:
chat/background/task.py:447-697,1322-1440 and SURVEY.md §3.2): dedup ->
incident linking -> aurora_status='running' + investigation_started_at
(the MTTD stamp) -> lifecycle event -> on_incident actions -> the headless
agent workflow -> status update, severity, summary, recommendations,
citations, postmortem, notifications, SSE.
"""
from __future__ import annotations

import json
import time
from typing import Any, Dict, Optional

from ..agent.state import State
from ..agent.workflow import Workflow
from .postrca import build_postmortem, generate_recommendations, save_citations
from .summarization import (alert_only_summary, determine_severity,
                            generate_incident_summary)


def build_rca_prompt(source: str, title: str, payload: dict,
                     prior_rcas: Optional[list] = None,
                     prediscovery: str = "") -> str:
    """reference: rca_prompt_builder.py:336 — alert-field extraction +
    prediscovery context injection + 'Aurora Learn' similar-good-RCA
    injection."""
    lines = [
        f"A {source} alert fired. Investigate it end-to-end and produce an RCA.",
        f"Title: {title}",
        f"Alert payload: {json.dumps(payload)[:4000]}",
    ]
    if prediscovery:
        lines.append(f"Known infrastructure (prediscovery): {prediscovery[:2000]}")
    if prior_rcas:
        lines.append("Similar past investigations (for reference):")
        for r in prior_rcas[:2]:
            lines.append(f"- {r[:500]}")
    lines.append("Investigate now. Use tools to gather evidence; finish with "
                 "final_answer containing the RCA.")
    return "\n".join(lines)


async def run_background_chat(rt, args: Dict[str, Any]) -> dict:
    org_id = args["org_id"]
    incident_id = args["incident_id"]
    user_id = args.get("user_id")
    store = rt.org(org_id)
    inc = store.get_incident(incident_id)
    if inc is None:
        return {"error": "incident not found"}
    if inc["aurora_status"] == "running":
        return {"skipped": "already running"}  # dedup (task.py:488-499)

    session_id = inc.get("aurora_chat_session_id") or store.create_session(
        user_id=user_id, is_background=True, incident_id=incident_id)
    t_pickup = time.time()
    store.update_incident(
        incident_id, aurora_status="running", status="investigating",
        aurora_chat_session_id=session_id,
        investigation_started_at=t_pickup)  # MTTD stamp (task.py:620-626)
    store.lifecycle(incident_id, "rca_started", {"session_id": session_id})
    rt.events.publish(org_id, {"type": "incident_update", "incident_id": incident_id,
                               "aurora_status": "running"})
    # billing/entitlement hook gate (reference: task.py:677-697)
    from ..utils.hooks import fire
    if not await fire("before_llm_call", org_id=org_id, user_id=user_id,
                      purpose="background_rca", default=True):
        store.update_incident(incident_id, aurora_status="failed")
        store.lifecycle(incident_id, "rca_denied", {"reason": "hook gate"})
        return {"error": "before_llm_call hook denied"}
    # on_incident actions (reference: services/actions/executor.py:111)
    rt.queue.enqueue("dispatch_on_incident_actions",
                     {"org_id": org_id, "incident_id": incident_id})
    from .notifications import notify_investigation_started
    await notify_investigation_started(rt, org_id, incident_id)

    payload = json.loads(inc.get("payload") or "{}")
    from .prediscovery import prediscovery_context
    # "Aurora Learn": inject summaries of similar completed RCAs
    prior = [i["summary"] for i in store.list_incidents(limit=30)
             if i["id"] != incident_id and i.get("summary")
             and i["aurora_status"] == "completed"][:2]
    prompt = args.get("prompt") or build_rca_prompt(
        inc["source"], inc["title"], payload, prior_rcas=prior,
        prediscovery=prediscovery_context(store))
    llm = rt.llm(org_id)
    ctx = rt.tool_ctx(org_id, user_id=user_id, session_id=session_id,
                      incident_id=incident_id, depth=1)  # depth=1 blocks nested RCA
    state = State(question=prompt, user_id=user_id, org_id=org_id,
                  session_id=session_id, incident_id=incident_id,
                  incident_start_time=inc["created_at"], is_background=True)
    final = ""
    error = None
    try:
        if rt.settings.orchestrator_enabled:
            # multi-agent path (reference: workflow.py:171-205 with
            # ORCHESTRATOR_ENABLED=true)
            from ..agent.orchestrator import Orchestrator
            orch = Orchestrator(llm, ctx,
                                sub_agent_turns=int(args.get("max_turns", 0))
                                or rt.settings.rca_max_turns or 6)
            result = await orch.run(state)
            final = result.final
        else:
            wf = Workflow(llm, ctx, max_turns=int(args.get("max_turns", 0))
                          or rt.settings.rca_max_turns or None,
                          tool_subset=args.get("tool_subset"))
            async for ev in wf.stream(state):
                if ev["type"] == "final":
                    final = ev["content"]
                elif ev["type"] == "blocked":
                    error = f"guardrail blocked: {ev['reason']}"
                elif ev["type"] == "error":
                    error = ev["error"]
    except Exception as e:  # noqa: BLE001
        msg = str(e)
        if "engine step failed" in msg or "engine shutdown" in msg:
            meta = args.get("_task_meta") or {}
            if meta.get("attempt", 1) < meta.get("max_attempts", 3):
                # transient engine fault: re-raise so the queue retries with
                # backoff (semantic failures below stay terminal). Status
                # must leave 'running' FIRST or the retry dedups against our
                # own crashed attempt and the incident strands (found by the
                # fault-injection test; Celery acks-late redelivers fresh).
                store.update_incident(incident_id, aurora_status="pending")
                store.lifecycle(incident_id, "rca_engine_fault", {"error": msg})
                raise
            # LAST attempt: fail terminally so the incident is never left
            # 'pending' with no retry coming (the stale-investigation
            # reaper is the backstop, not the primary path)
            store.lifecycle(incident_id, "rca_engine_fault_final", {"error": msg})
            error = f"engine fault (retries exhausted): {e}"
        else:
            error = f"workflow failure: {e}"

    # ---- post-RCA stage ----
    if error and not final:
        summary = alert_only_summary(inc["title"], payload)
        store.update_incident(incident_id, aurora_status="failed", summary=summary)
        store.lifecycle(incident_id, "rca_failed", {"error": error})
        rt.events.publish(org_id, {"type": "incident_update",
                                   "incident_id": incident_id,
                                   "aurora_status": "failed"})
        return {"error": error}

    # summary and recommendations are independent LLM passes: run them
    # concurrently so the post-RCA tail keeps the engine's decode batch
    # occupied instead of serializing 4 calls per incident
    import asyncio as _aio
    summary_task = _aio.create_task(
        generate_incident_summary(llm, state.messages, session_id))
    summary_task.add_done_callback(
        lambda t: None if t.cancelled() else t.exception())
    n_cits = save_citations(store, incident_id, state.messages)
    suggestions = await generate_recommendations(llm, store, incident_id,
                                                 state.messages, ctx=ctx)
    try:
        summary = await summary_task
    except Exception:
        summary = alert_only_summary(inc["title"], payload)
    severity = determine_severity(summary, inc.get("severity"))
    store.update_incident(incident_id, aurora_status="completed",
                          summary=summary, severity=severity,
                          root_cause=_extract_root_cause(summary))
    store.lifecycle(incident_id, "rca_completed",
                    {"citations": n_cits, "suggestions": len(suggestions)})
    # auto postmortem (system action)
    pm = build_postmortem(store.get_incident(incident_id), summary,
                          store.citations(incident_id), suggestions,
                          store.lifecycle_events(incident_id))
    store.save_postmortem(incident_id, pm)
    rt.events.publish(org_id, {"type": "incident_update", "incident_id": incident_id,
                               "aurora_status": "completed",
                               "severity": severity})
    from .visualization import publish_visualizations
    publish_visualizations(rt, org_id, incident_id, state.messages)
    from .notifications import notify_investigation_completed
    await notify_investigation_completed(rt, org_id, incident_id)
    await fire("report_usage", org_id=org_id, incident_id=incident_id, default=None)
    return {"ok": True, "summary_chars": len(summary), "citations": n_cits,
            "time_to_rca_s": time.time() - inc["created_at"]}


def _extract_root_cause(summary: str) -> str:
    for line in summary.splitlines():
        low = line.strip().lower()
        if low.startswith("root cause"):
            return line.strip()[:500]
    return ""


async def trigger_delayed_rca(rt, args: Dict[str, Any]) -> dict:
    """The second Celery hop of the webhook path (reference:
    routes/pagerduty/tasks.py:235-431): dedup via the session link, rate
    limit, build prompt, create session, dispatch run_background_chat."""
    org_id = args["org_id"]
    incident_id = args["incident_id"]
    store = rt.org(org_id)
    inc = store.get_incident(incident_id)
    if inc is None:
        return {"error": "incident gone"}
    if inc.get("aurora_chat_session_id"):
        return {"skipped": "rca already dispatched"}
    # admission control (reference: is_background_chat_allowed,
    # task.py:336): when too many investigations are already running for
    # this org, DEFER (re-enqueue with countdown) rather than drop — a
    # concurrency cap, not the per-user cooldown (that one gates the
    # agent-initiated trigger_rca TOOL only)
    running = sum(1 for r in store.list_incidents(limit=500)
                  if r.get("aurora_status") == "running")
    if running >= rt.settings.max_concurrent_rca:
        defers = int(args.get("defer_count", 0))
        if defers >= 60:  # ~30 min of deferral -> give up loudly
            store.lifecycle(incident_id, "rca_defer_exhausted",
                            {"running": running})
            store.update_incident(incident_id, aurora_status="failed")
            return {"skipped": "defer exhausted"}
        store.lifecycle(incident_id, "rca_deferred", {"running": running})
        rt.queue.enqueue("trigger_delayed_rca",
                         dict(args, defer_count=defers + 1),
                         countdown_s=30.0)
        return {"deferred": True}
    tid = rt.queue.enqueue(
        "run_background_chat",
        {"org_id": org_id, "incident_id": incident_id,
         "user_id": args.get("user_id"),
         "max_turns": args.get("max_turns", 0),
         "tool_subset": args.get("tool_subset")},
        dedup_key=f"rca:{incident_id}")
    if tid:
        store.update_incident(incident_id, rca_task_id=tid)
    return {"task_id": tid}


async def cleanup_stale_investigations(rt, args: Dict[str, Any]) -> dict:
    """Beat: mark investigations whose heartbeat (incidents.updated_at,
    touched on every thought save) went stale as failed (reference:
    task.py:2390,2442,2350)."""
    stale = rt.admin.stale_running_incidents(older_than_s=900.0)
    for inc in stale:
        store = rt.org(inc["org_id"])
        store.update_incident(inc["id"], aurora_status="failed")
        store.lifecycle(inc["id"], "rca_reaped", {"reason": "stale heartbeat"})
        rt.events.publish(inc["org_id"], {
            "type": "incident_update", "incident_id": inc["id"],
            "aurora_status": "failed"})
    return {"reaped": len(stale)}


async def dispatch_on_incident_actions(rt, args: Dict[str, Any]) -> dict:
    """reference: services/actions/executor.py:16,111"""
    from .actions import run_incident_actions
    return await run_incident_actions(rt, args["org_id"], args["incident_id"])


async def cleanup_sandboxes(rt, args: Dict[str, Any]) -> dict:
    """Beat: remove idle terminal sandboxes (reference:
    terminal_pod_cleanup.py:85, every 10 min)."""
    from ..agent.tools.sandbox import sandbox_manager
    return {"removed": sandbox_manager().cleanup_idle()}


async def action_scheduler(rt, args: Dict[str, Any]) -> dict:
    """Per-minute scheduled-action beat (reference: scheduler.py:38)."""
    from .actions import run_scheduled_actions
    return await run_scheduled_actions(rt)


def register_tasks(rt) -> None:
    q = rt.queue
    q.register("run_background_chat", lambda a: run_background_chat(rt, a))
    q.register("trigger_delayed_rca", lambda a: trigger_delayed_rca(rt, a))
    q.register("cleanup_stale_investigations",
               lambda a: cleanup_stale_investigations(rt, a))
    q.register("dispatch_on_incident_actions",
               lambda a: dispatch_on_incident_actions(rt, a))
    q.register("action_scheduler", lambda a: action_scheduler(rt, a))
    q.register("cleanup_sandboxes", lambda a: cleanup_sandboxes(rt, a))
    from .change_gating import investigate_pr
    from .prediscovery import run_prediscovery
    from .chat_integrations import handle_chat_command
    q.register("investigate_pr", lambda a: investigate_pr(rt, a))
    q.register("run_prediscovery", lambda a: run_prediscovery(rt, a))
    q.register("handle_chat_command", lambda a: handle_chat_command(rt, a))
