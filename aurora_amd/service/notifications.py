"""Notification dispatcher (reference: utils/notifications/dispatcher.py —
email/Slack/Google Chat investigation-started/completed).  With no egress,
deliveries degrade to the event bus + an audit row; the provider-webhook
POST path is implemented and used when a sink URL is configured."""
from __future__ import annotations

import json


async def dispatch_notification(rt, org_id: str, message: str,
                                incident_id: str = "",
                                kind: str = "info") -> None:
    store = rt.org(org_id)
    store.audit("notification", json.dumps(
        {"message": message[:500], "incident_id": incident_id, "kind": kind}))
    rt.events.publish(org_id, {"type": "notification", "message": message,
                               "incident_id": incident_id, "kind": kind})
    sinks = store.get_token_ref("system", "notification_webhook")
    if sinks:
        try:
            import httpx
            async with httpx.AsyncClient(timeout=5.0) as client:
                await client.post(sinks, json={"text": message,
                                               "incident_id": incident_id})
        except Exception as e:  # noqa: BLE001 - notification failure is non-fatal
            store.audit("notification_error", str(e)[:300])


async def notify_investigation_started(rt, org_id: str, incident_id: str) -> None:
    inc = rt.org(org_id).get_incident(incident_id)
    if inc:
        await dispatch_notification(
            rt, org_id, f"Aurora started investigating: {inc['title']}",
            incident_id, "investigation_started")


async def notify_investigation_completed(rt, org_id: str, incident_id: str) -> None:
    inc = rt.org(org_id).get_incident(incident_id)
    if inc:
        await dispatch_notification(
            rt, org_id,
            f"Investigation complete ({inc.get('severity')}): {inc['title']}",
            incident_id, "investigation_completed")
