"""Two-way chat integrations: Slack and Google Chat.

Rebuilds the reference's chat-platform bridges (reference: events routes +
_send_response_to_slack / _send_response_to_google_chat,
chat/background/task.py:2016,2160): an @-mention or slash command in the
chat platform becomes an agent query; the final answer is posted back via
the platform's response/webhook URL (degrades to the event bus + audit
when there is no egress)."""
from __future__ import annotations

import json
from typing import Any, Dict, Optional


def normalize_chat_event(platform: str, payload: dict) -> Optional[dict]:
    if platform == "slack":
        ev = payload.get("event") or {}
        if payload.get("type") == "url_verification":
            return {"challenge": payload.get("challenge", "")}
        text = ev.get("text") or payload.get("text", "")
        if not text:
            return None
        return {"platform": "slack", "text": text,
                "user": ev.get("user") or payload.get("user_id", ""),
                "channel": ev.get("channel") or payload.get("channel_id", ""),
                "response_url": payload.get("response_url", "")}
    if platform == "google_chat":
        msg = payload.get("message") or {}
        text = msg.get("argumentText") or msg.get("text", "")
        if not text:
            return None
        return {"platform": "google_chat", "text": text,
                "user": ((msg.get("sender") or {}).get("name", "")),
                "channel": ((payload.get("space") or {}).get("name", "")),
                "response_url": ""}
    return None


async def send_chat_response(rt, org_id: str, event: dict, text: str) -> None:
    store = rt.org(org_id)
    url = event.get("response_url", "")
    if url:
        try:
            import httpx
            async with httpx.AsyncClient(timeout=8.0) as client:
                await client.post(url, json={"text": text[:3500]})
            return
        except Exception as e:  # noqa: BLE001
            store.audit("chat_send_error", str(e)[:300])
    rt.events.publish(org_id, {"type": "chat_response",
                               "platform": event.get("platform"),
                               "channel": event.get("channel"),
                               "text": text[:3500]})
    store.audit("chat_response", json.dumps(
        {"platform": event.get("platform"), "chars": len(text)}))


async def handle_chat_command(rt, args: Dict[str, Any]) -> dict:
    """Task: run the agent for a chat-platform query, reply on the platform."""
    from ..agent.state import State
    from ..agent.workflow import Workflow

    org_id = args["org_id"]
    event = args["event"]
    store = rt.org(org_id)
    session_id = store.create_session(user_id=event.get("user") or None)
    ctx = rt.tool_ctx(org_id, user_id=args.get("user_id"),
                      session_id=session_id, mode=args.get("mode", "ask"))
    wf = Workflow(rt.llm(org_id), ctx,
                  max_turns=int(args.get("max_turns", 0)) or 8)
    state = State(question=event["text"], session_id=session_id,
                  org_id=org_id, mode=args.get("mode", "ask"))
    final = ""
    async for ev in wf.stream(state):
        if ev["type"] == "final":
            final = ev["content"]
        elif ev["type"] == "blocked":
            final = f"(blocked by guardrail: {ev['reason']})"
    await send_chat_response(rt, org_id, event, final or "(no answer)")
    return {"chars": len(final)}
