"""Webhook robustness fuzz: malformed / adversarial payloads must never
crash ingestion — every request returns an HTTP status (not a 500 from an
unhandled exception), and valid-but-weird payloads still create
incidents (service/webhooks.py per-source extractors)."""
import json

import pytest
from hypothesis import given, settings, strategies as st

from fastapi.testclient import TestClient

from aurora_amd.engine.engine import EngineConfig
from aurora_amd.engine.server import AsyncEngineServer
from aurora_amd.service.app import create_app
from aurora_amd.service.runtime import Runtime
from aurora_amd.service.webhooks import sources
from aurora_amd.utils.config import Settings


@pytest.fixture(scope="module")
def env():
    cfg = Settings(db_path=":memory:", secrets_path="/tmp/fz_secrets.json",
                   rca_max_turns=1)
    server = AsyncEngineServer(EngineConfig(model="tiny-llama", device="cpu"))
    rt = Runtime(cfg, engine_server=server)
    org = rt.admin.create_org("fuzz")
    rt.admin.create_user(org, "f@f.io", user_id="fz")
    app = create_app(rt)
    with TestClient(app, raise_server_exceptions=False) as client:
        yield client, org
    server.shutdown()


json_soup = st.recursive(
    st.none() | st.booleans() | st.integers() | st.text(max_size=30),
    lambda c: st.lists(c, max_size=3) | st.dictionaries(st.text(max_size=8), c, max_size=3),
    max_leaves=8)


@settings(max_examples=60, deadline=None)
@given(json_soup)
def test_webhook_payload_soup_never_500(env, payload):
    client, org = env
    for source in list(sources())[:6]:
        r = client.post(f"/webhook/{source}/{org}", json=payload)
        assert r.status_code < 500, (source, r.status_code, r.text[:200])


def test_webhook_non_json_and_huge_fields(env):
    client, org = env
    r = client.post(f"/webhook/pagerduty/{org}", content=b"\x00not json",
                    headers={"content-type": "application/json"})
    assert r.status_code < 500
    r2 = client.post(f"/webhook/pagerduty/{org}", json={
        "event": {"eventType": "incident.triggered",
                  "data": {"id": "x" * 100000, "title": "y" * 200000}}})
    assert r2.status_code < 500


def test_webhook_unknown_source_and_org(env):
    client, org = env
    assert client.post(f"/webhook/definitely-not-a-source/{org}",
                       json={}).status_code in (400, 404)
    assert client.post("/webhook/pagerduty/org-does-not-exist",
                       json={}).status_code in (400, 403, 404)
