"""Tests for the lighter background subsystems: prediscovery context,
PR change-gating normalization/dispatch, and notification dispatch."""
import asyncio

import pytest

from aurora_amd.engine.engine import EngineConfig
from aurora_amd.engine.server import AsyncEngineServer
from aurora_amd.service.runtime import Runtime
from aurora_amd.utils.config import Settings


@pytest.fixture(scope="module")
def rt():
    server = AsyncEngineServer(EngineConfig(model="tiny-llama", device="cpu"))
    r = Runtime(Settings(db_path=":memory:", secrets_path="/tmp/bg_sec.json",
                         rca_max_turns=1), engine_server=server)
    yield r
    server.shutdown()


@pytest.fixture(scope="module")
def org(rt):
    o = rt.admin.create_org("bg")
    rt.admin.create_user(o, "bg@bg.io", user_id="bg-user")
    return o


def test_pr_event_normalization():
    from aurora_amd.service.change_gating import normalize_pr_event
    gh = normalize_pr_event("github", {
        "action": "opened",
        "pull_request": {"number": 7, "title": "fix: retry logic",
                         "html_url": "https://x/pr/7",
                         "head": {"ref": "fix"}, "base": {"ref": "main"},
                         "user": {"login": "dev"}},
        "repository": {"full_name": "acme/api"}})
    assert gh and gh["number"] == 7 and gh["repo"] == "acme/api"
    assert normalize_pr_event("github", {"action": "labeled"}) is None
    assert normalize_pr_event("github", {"nonsense": True}) is None
    bb = normalize_pr_event("bitbucket", {
        "pullrequest": {"id": 3, "title": "chore", "links": {},
                        "source": {"branch": {"name": "b"}},
                        "destination": {"branch": {"name": "main"}}},
        "repository": {"full_name": "acme/infra"}})
    assert bb and bb["number"] == 3


def test_pr_webhook_dispatches_review_task(rt, org):
    from aurora_amd.service.change_gating import process_pr_webhook
    tid = process_pr_webhook(rt, org, "github", {
        "action": "opened",
        "pull_request": {"number": 11, "title": "risky migration",
                         "html_url": "https://x/pr/11",
                         "head": {"ref": "mig"}, "base": {"ref": "main"},
                         "user": {"login": "dev"}},
        "repository": {"full_name": "acme/api"}})
    assert tid
    # ignored action -> no task
    assert process_pr_webhook(rt, org, "github", {"action": "labeled"}) is None


def test_prediscovery_context_roundtrip(rt, org):
    from aurora_amd.service.prediscovery import prediscovery_context
    store = rt.org(org)
    ctx = prediscovery_context(store)
    assert isinstance(ctx, str)


def test_notification_dispatch_records(rt, org):
    from aurora_amd.service.notifications import (
        notify_investigation_completed, notify_investigation_started)
    store = rt.org(org)
    iid = store.create_incident(title="notif probe", source="datadog",
                                payload={})
    asyncio.run(notify_investigation_started(rt, org, iid))
    asyncio.run(notify_investigation_completed(rt, org, iid))
    rows = store.notifications() if hasattr(store, "notifications") else None
    # at minimum the calls must not raise; if persisted, they reference iid
    if rows:
        assert any(iid in str(r) for r in rows)


def test_chat_command_end_to_end(rt, org):
    """Slack-style chat query -> agent run -> platform reply recorded
    (service/chat_integrations.py handle_chat_command)."""
    from aurora_amd.service import chat_integrations as ci

    sent = []

    async def fake_send(rt_, org_id, event, text):
        sent.append((event.get("channel"), text))

    orig = ci.send_chat_response
    ci.send_chat_response = fake_send
    try:
        out = asyncio.run(ci.handle_chat_command(rt, {
            "org_id": org, "user_id": "bg-user", "mode": "ask",
            "max_turns": 1,
            "event": {"platform": "slack", "text": "what incidents are open?",
                      "channel": "C123", "user": "U1"}}))
        assert sent and sent[0][0] == "C123"
        assert isinstance(out.get("chars"), int)
    finally:
        ci.send_chat_response = orig


def test_serve_module_builds_app(rt):
    """serve.py's app assembly imports and builds without a server."""
    from aurora_amd.service.app import create_app
    app = create_app(rt)
    paths = {r.path for r in app.routes}
    assert {"/health", "/ui", "/incidents", "/metrics/prometheus"} <= paths


def test_rca_admission_defers_over_cap(rt, org):
    """Over the MAX_CONCURRENT_RCA cap, new webhook RCAs defer (re-enqueue
    with countdown) instead of dropping (reference
    is_background_chat_allowed, task.py:336)."""
    from aurora_amd.service.worker import trigger_delayed_rca
    store = rt.org(org)
    old = rt.settings.max_concurrent_rca
    rt.settings.max_concurrent_rca = 1
    try:
        busy = store.create_incident(title="busy", source="datadog", payload={})
        store.update_incident(busy, aurora_status="running")
        iid = store.create_incident(title="queued", source="datadog", payload={})
        out = asyncio.run(trigger_delayed_rca(
            rt, {"org_id": org, "incident_id": iid}))
        assert out.get("deferred") is True
        evs = [e for e in store.lifecycle_events(iid) if e["event"] == "rca_deferred"]
        assert evs
        # deferral must re-enqueue even when the running count is unchanged
        out2 = asyncio.run(trigger_delayed_rca(
            rt, {"org_id": org, "incident_id": iid, "defer_count": 1}))
        assert out2.get("deferred") is True
        # exhaustion fails loudly instead of deferring forever
        out3 = asyncio.run(trigger_delayed_rca(
            rt, {"org_id": org, "incident_id": iid, "defer_count": 60}))
        assert out3.get("skipped")
        assert store.get_incident(iid)["aurora_status"] == "failed"
    finally:
        rt.settings.max_concurrent_rca = old


def test_alert_gap_system_action():
    """alert_gap finds graph services with no alert-sourced incidents
    (reference: system alert-gap action)."""
    import asyncio
    import json as _json
    from aurora_amd.engine.engine import EngineConfig
    from aurora_amd.engine.server import AsyncEngineServer
    from aurora_amd.service.actions import _execute
    from aurora_amd.service.runtime import Runtime
    from aurora_amd.utils.config import Settings

    srv = AsyncEngineServer(EngineConfig(model="tiny-llama", device="cpu"))
    rt = Runtime(Settings(db_path=":memory:",
                          secrets_path="/tmp/ag_sec.json"),
                 engine_server=srv)
    oid = rt.admin.create_org("gap")
    store = rt.org(oid)
    g = rt.graph(oid)
    for svc in ("api", "db", "shadow-svc"):
        g.upsert_service(svc)
    store.create_incident("api 5xx", "pagerduty", {"service": "api"})

    async def main():
        await _execute(rt, oid, {"id": "a1", "name": "gap-check",
                                 "config": _json.dumps({"kind": "alert_gap"})})
    asyncio.run(main())
    rep = _json.loads(store.get_artifact("alert_gap_report")["content"])
    assert "api" in rep["covered"]
    assert "shadow-svc" in rep["uncovered_services"]
    assert "db" in rep["uncovered_services"]
    srv.shutdown()
