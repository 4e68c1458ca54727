"""Connector OAuth flows against fixtures (reference: connectors/<name>
OAuth plumbing; VERDICT r1 missing #4): authorize-URL construction,
one-shot CSRF state, code->token exchange, refresh merge, route wiring."""
import asyncio
import time
import urllib.parse

import pytest

from aurora_amd.connectors.clients.base import ProviderAPIError
from aurora_amd.connectors.oauth import (StateCache, authorize_url,
                                         build_token_exchange,
                                         build_token_refresh, exchange_code,
                                         needs_refresh, parse_token_response,
                                         refresh_credentials)


def test_authorize_url_shapes():
    u = authorize_url("github", "cid", "https://app/cb", "st4te")
    q = dict(urllib.parse.parse_qsl(urllib.parse.urlsplit(u).query))
    assert u.startswith("https://github.com/login/oauth/authorize?")
    assert q["client_id"] == "cid" and q["state"] == "st4te"
    assert q["redirect_uri"] == "https://app/cb" and "repo" in q["scope"]
    u = authorize_url("gcp", "cid", "https://app/cb", "s")
    q = dict(urllib.parse.parse_qsl(urllib.parse.urlsplit(u).query))
    assert q["access_type"] == "offline"  # refresh tokens need consent
    with pytest.raises(ProviderAPIError):
        authorize_url("not-a-provider", "c", "r", "s")


def test_state_cache_one_shot_and_expiry():
    sc = StateCache(ttl_s=100)
    s = sc.issue("u1", "github")
    got = sc.consume(s)
    assert got["user_id"] == "u1" and got["provider"] == "github"
    assert sc.consume(s) is None  # one-shot
    s2 = sc.issue("u1", "slack")
    sc._states[s2]["ts"] = time.time() - 1000
    assert sc.consume(s2) is None  # expired


def test_token_exchange_and_parse():
    req = build_token_exchange("github", "cid", "sec", "code123",
                               "https://app/cb")
    assert req.url == "https://github.com/login/oauth/access_token"
    assert req.json_body["grant_type"] == "authorization_code"
    assert req.json_body["code"] == "code123"
    creds = parse_token_response({"access_token": "at", "expires_in": 3600,
                                  "refresh_token": "rt", "scope": "repo"})
    assert creds["access_token"] == "at" and creds["refresh_token"] == "rt"
    assert creds["expires_at"] > time.time()
    # slack nests the user token
    creds = parse_token_response({"ok": True,
                                  "authed_user": {"access_token": "xoxp"}})
    assert creds["access_token"] == "xoxp"
    with pytest.raises(ProviderAPIError):
        parse_token_response({"error": "bad_verification_code"})
    with pytest.raises(ProviderAPIError):
        parse_token_response({})


def test_refresh_merge_keeps_refresh_token():
    async def main():
        async def t(req):
            assert req.json_body["grant_type"] == "refresh_token"
            return {"access_token": "new", "expires_in": 60}
        creds = {"access_token": "old", "refresh_token": "keepme",
                 "cloud_id": "xyz"}
        merged = await refresh_credentials(t, "jira", "c", "s", creds)
        assert merged["access_token"] == "new"
        assert merged["refresh_token"] == "keepme"  # not dropped
        assert merged["cloud_id"] == "xyz"          # provider extras kept
        assert not needs_refresh(merged, skew_s=0)
        assert needs_refresh({"expires_at": time.time() - 1})
        with pytest.raises(ProviderAPIError):
            await refresh_credentials(t, "jira", "c", "s", {})
    asyncio.run(main())


def test_oauth_routes_end_to_end():
    """start -> callback against a fixture transport: token stored via the
    secrets backend, provider marked connected."""
    from fastapi.testclient import TestClient
    from aurora_amd.engine.engine import EngineConfig
    from aurora_amd.engine.server import AsyncEngineServer
    from aurora_amd.service.app import create_app
    from aurora_amd.service.runtime import Runtime
    from aurora_amd.utils.config import Settings
    from aurora_amd.utils.secrets import reset_backend, secrets_backend

    import os
    import tempfile
    with tempfile.TemporaryDirectory() as td:
        os.environ["AURORA_SECRETS"] = os.path.join(td, "sec.json")
        reset_backend()
        try:
            cfg = Settings(db_path=":memory:",
                           secrets_path=os.path.join(td, "sec.json"))
            server = AsyncEngineServer(EngineConfig(model="tiny-llama",
                                                    device="cpu", seed=2))
            rt = Runtime(cfg, engine_server=server)
            oid = rt.admin.create_org("oa")
            uid = rt.admin.create_user(oid, "o@a.io", user_id="user-oa")

            async def fixture(req):
                assert req.url.endswith("access_token")
                assert req.json_body["code"] == "authcode"
                return {"access_token": "gh-token-1", "scope": "repo"}
            rt.provider_transport = fixture
            # register the OAuth app credentials
            secrets_backend().put("oauth_app:github",
                                  {"client_id": "cid", "client_secret": "s",
                                   "redirect_uri": "https://app/cb"})
            store = rt.org(oid)
            store.store_tokens("system", "oauth_app:github",
                               "oauth_app:github")
            app = create_app(rt)
            with TestClient(app) as client:
                H = {"X-User-Id": uid}
                r = client.get("/connectors/github/oauth/start", headers=H)
                assert r.status_code == 200, r.text
                state = r.json()["state"]
                assert "github.com/login/oauth" in r.json()["authorize_url"]
                r = client.get("/connectors/oauth/callback",
                               params={"code": "authcode", "state": state},
                               headers=H)
                assert r.status_code == 200, r.text
                assert "github" in store.connected_providers(uid)
                from aurora_amd.connectors.registry import get_credentials
                creds = get_credentials(store, uid, "github")
                assert creds["access_token"] == "gh-token-1"
                # replayed state fails closed
                r = client.get("/connectors/oauth/callback",
                               params={"code": "x", "state": state},
                               headers=H)
                assert r.status_code == 403
            server.shutdown()
        finally:
            os.environ.pop("AURORA_SECRETS", None)
            reset_backend()
