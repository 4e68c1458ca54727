"""HTTP API (FastAPI): the reference's Flask surface rebuilt.

Route groups mirror reference blueprints (reference: main_compute.py:343-587
registrations; routes/*): webhooks per source, incidents CRUD + SSE,
monitor (waterfall/fleet), metrics (MTTD/pickup), postmortems, artifacts,
findings, llm usage, audit, command policies, knowledge base, graph,
connectors, MCP-dispatchable tool surface, health.

Auth model: X-User-Id + X-Org-Id headers resolved/verified against the user
table (the reference's stateless-auth + user<->org binding middleware,
main_compute.py:299); INTERNAL_API_SECRET header for service-to-service
calls (AGENTS.md:96).
"""
from __future__ import annotations

import asyncio
import contextvars
import hmac
import json
import time
from typing import Optional

from fastapi import Depends, FastAPI, Header, HTTPException, Request
from fastapi.responses import StreamingResponse

from ..utils.auth_tokens import mint_token, verify_token
from ..utils.config import settings
from .runtime import Runtime
from .store import OrgStore
from .webhooks import process_webhook, sources, verify_signature


def create_app(rt: Runtime) -> FastAPI:
    app = FastAPI(title="aurora-mi355x", version="0.1.0")
    app.state.rt = rt

    @app.on_event("startup")
    async def _startup():
        rt.start()

    @app.on_event("shutdown")
    async def _shutdown():
        await rt.stop()

    # ---------------------------------------------------------------- auth
    # RBAC: 3 roles (reference: Casbin model rbac_model.conf + 
    # require_permission decorators, utils/auth/rbac_decorators.py).
    ROLE_PERMS = {
        "viewer": {"read"},
        "member": {"read", "write", "investigate", "connect"},
        "admin": {"read", "write", "investigate", "connect", "admin"},
    }

    # identity resolved by auth() for the current request: "__internal__"
    # for service-secret calls, else the authenticated user id. _require
    # checks THIS, never a client-supplied header, so omitting X-User-Id
    # can't skip RBAC in token mode.
    _ident: contextvars.ContextVar[Optional[str]] = \
        contextvars.ContextVar("auth_ident", default=None)

    def _require(verb: str, x_user_id: Optional[str] = None) -> None:
        ident = _ident.get()
        if ident == "__internal__":
            return  # internal-secret calls carry full permissions
        uid = ident or x_user_id
        if not uid:
            raise HTTPException(401, "authentication required")
        user = rt.admin.get_user(uid)
        role = (user or {}).get("role", "viewer")
        if verb not in ROLE_PERMS.get(role, {"read"}):
            raise HTTPException(403, f"role '{role}' lacks '{verb}' permission")

    async def auth(x_user_id: Optional[str] = Header(None),
                   x_org_id: Optional[str] = Header(None),
                   x_internal_api_secret: Optional[str] = Header(None),
                   authorization: Optional[str] = Header(None)) -> OrgStore:
        # async so the ContextVar set here is visible to the route handler
        # (sync dependencies run in a copied threadpool context)
        s = rt.settings
        if s.internal_api_secret and x_internal_api_secret is not None:
            if not hmac.compare_digest(x_internal_api_secret,
                                       s.internal_api_secret):
                raise HTTPException(401, "bad internal API secret")
            if x_org_id:
                _ident.set("__internal__")
                return rt.org(x_org_id)
            raise HTTPException(400, "internal call requires X-Org-Id")
        if s.internal_api_secret:
            # authenticated mode: identity comes ONLY from an HMAC-signed
            # bearer token (reference signs JWTs, main_chatbot.py:107-158);
            # a bare X-User-Id is never trusted when a secret is configured
            # (identity-header anti-spoofing).
            tok = (authorization or "")
            if tok.lower().startswith("bearer "):
                tok = tok[7:].strip()
            claims = verify_token(tok, s.internal_api_secret)
            if claims is None:
                raise HTTPException(401, "valid bearer token required")
            if x_user_id and x_user_id != claims["user_id"]:
                raise HTTPException(401, "X-User-Id does not match token")
            x_user_id = claims["user_id"]
        elif not x_user_id:
            raise HTTPException(401, "X-User-Id required")
        user = rt.admin.get_user(x_user_id)
        if user is None:
            raise HTTPException(401, "unknown user")
        # user<->org binding enforcement (main_compute.py:299): a client-sent
        # org id that mismatches the user's org is rejected, not trusted
        if x_org_id and x_org_id != user["org_id"]:
            raise HTTPException(403, "user is not a member of that org")
        _ident.set(x_user_id)
        return rt.org(user["org_id"])

    @app.post("/auth/token")
    async def issue_token(body: dict,
                          x_internal_api_secret: Optional[str] = Header(None)):
        """Mint a short-lived signed identity token for a user. Only the
        trusted frontend proxy (which holds INTERNAL_API_SECRET, the
        reference's service-to-service trust anchor, AGENTS.md:96) may mint."""
        s = rt.settings
        if not s.internal_api_secret:
            raise HTTPException(404, "token auth not enabled")
        if not (x_internal_api_secret and hmac.compare_digest(
                x_internal_api_secret, s.internal_api_secret)):
            raise HTTPException(401, "internal API secret required")
        uid = body.get("user_id", "")
        if rt.admin.get_user(uid) is None:
            raise HTTPException(404, "unknown user")
        ttl = min(int(body.get("ttl_s", 600)), 86400)
        return {"token": mint_token(uid, s.internal_api_secret, ttl_s=ttl),
                "ttl_s": ttl}

    # ------------------------------------------------------------- webhooks

    def _require_org(org_id: str) -> None:
        """Webhook URLs embed the org id; an unknown org must 404 instead
        of silently materialising tenant state (org<->binding middleware,
        reference: main_compute.py:299)."""
        if not any(o["id"] == org_id for o in rt.admin.orgs()):
            raise HTTPException(404, "unknown org")

    async def _json_object(request: Request) -> dict:
        body = await request.body()
        try:
            payload = json.loads(body or b"{}")
        except (json.JSONDecodeError, UnicodeDecodeError):
            raise HTTPException(400, "invalid JSON") from None
        if not isinstance(payload, dict):
            raise HTTPException(400, "payload must be a JSON object")
        return payload

    @app.post("/webhook/{source}/{org_id}")
    async def webhook(source: str, org_id: str, request: Request,
                      user_id: Optional[str] = None):
        if source not in sources():
            raise HTTPException(404, f"unknown source; known: {sources()}")
        _require_org(org_id)
        body = await request.body()
        store = rt.org(org_id)
        secret = store.get_token_ref("system", f"webhook_secret:{source}")
        if not secret:
            if settings().require_webhook_secrets:
                store.audit("webhook_rejected_unsigned", source)
                raise HTTPException(403, "webhook secret required")
            store.audit("webhook_accepted_unsigned", source)
        elif not verify_signature(source, body, dict(request.headers), secret):
            store.audit("webhook_bad_signature", source)
            raise HTTPException(403, "bad signature")
        try:
            payload = json.loads(body or b"{}")
        except (json.JSONDecodeError, UnicodeDecodeError):
            raise HTTPException(400, "invalid JSON") from None
        if not isinstance(payload, dict):
            raise HTTPException(400, "payload must be a JSON object")
        try:
            iid, created = process_webhook(rt, org_id, source, payload,
                                           user_id=user_id)
        except (KeyError, TypeError, AttributeError, ValueError) as e:
            raise HTTPException(400, f"unprocessable alert payload: {e}") from None
        return {"incident_id": iid, "created": created}

    @app.post("/webhook/pr/{provider}/{org_id}")
    async def pr_webhook(provider: str, org_id: str, request: Request):
        # change gating (reference: tasks/github_webhook_tasks.py:1090)
        if provider not in ("github", "bitbucket"):
            raise HTTPException(404, "provider must be github|bitbucket")
        _require_org(org_id)
        body = await request.body()
        store = rt.org(org_id)
        secret = store.get_token_ref("system", f"webhook_secret:{provider}")
        from .webhooks import verify_signature as _vs
        if not secret:
            if settings().require_webhook_secrets:
                store.audit("webhook_rejected_unsigned", provider)
                raise HTTPException(403, "webhook secret required")
            store.audit("webhook_accepted_unsigned", provider)
        elif not _vs(provider, body, dict(request.headers), secret):
            raise HTTPException(403, "bad signature")
        from .change_gating import process_pr_webhook
        try:
            payload = json.loads(body or b"{}")
        except (json.JSONDecodeError, UnicodeDecodeError):
            raise HTTPException(400, "invalid JSON") from None
        if not isinstance(payload, dict):
            raise HTTPException(400, "payload must be a JSON object")
        tid = process_pr_webhook(rt, org_id, provider, payload)
        return {"task_id": tid}

    @app.post("/chat-events/{platform}/{org_id}")
    async def chat_events(platform: str, org_id: str, request: Request):
        # Slack / Google Chat two-way bridge
        if platform not in ("slack", "google_chat"):
            raise HTTPException(404, "platform must be slack|google_chat")
        _require_org(org_id)
        from .chat_integrations import normalize_chat_event
        payload = await _json_object(request)
        ev = normalize_chat_event(platform, payload)
        if ev is None:
            return {"ok": True}
        if "challenge" in ev:  # Slack URL verification handshake
            return {"challenge": ev["challenge"]}
        tid = rt.queue.enqueue("handle_chat_command",
                               {"org_id": org_id, "event": ev})
        return {"task_id": tid}

    @app.get("/incidents/{iid}/visualizations")
    async def visualizations(iid: str, store: OrgStore = Depends(auth)):
        row = store.get_artifact(f"visualizations:{iid}")
        return json.loads(row["content"]) if row else []

    # ------------------------------------------------------------ incidents

    @app.get("/incidents")
    async def list_incidents(status: Optional[str] = None, limit: int = 100,
                             store: OrgStore = Depends(auth)):
        return store.list_incidents(status, limit=max(1, min(limit, 1000)))

    @app.get("/incidents/{iid}")
    async def get_incident(iid: str, store: OrgStore = Depends(auth)):
        inc = store.get_incident(iid)
        if inc is None:
            raise HTTPException(404, "no such incident")
        return inc

    @app.post("/incidents")
    async def create_incident(body: dict, store: OrgStore = Depends(auth),
                              x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        iid = store.create_incident(
            title=body.get("title", "manual incident"),
            source=body.get("source", "manual"),
            payload=body.get("payload", {}))
        return {"incident_id": iid}

    @app.post("/incidents/{iid}/investigate")
    async def investigate(iid: str, store: OrgStore = Depends(auth),
                          x_user_id: Optional[str] = Header(None)):
        _require("investigate", x_user_id)
        if store.get_incident(iid) is None:
            raise HTTPException(404, "no such incident")
        tid = rt.queue.enqueue("run_background_chat",
                               {"org_id": store.org_id, "incident_id": iid},
                               dedup_key=f"rca:{iid}:{int(time.time())}")
        return {"task_id": tid}

    @app.post("/incidents/{iid}/cancel")
    async def cancel_rca(iid: str, store: OrgStore = Depends(auth),
                         x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        inc = store.get_incident(iid)
        if inc is None:
            raise HTTPException(404, "no such incident")
        ok = rt.queue.cancel(inc.get("rca_task_id") or "")
        return {"cancelled": ok}

    @app.post("/incidents/{iid}/resolve")
    async def resolve(iid: str, store: OrgStore = Depends(auth),
                      x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        store.update_incident(iid, status="resolved", resolved_at=time.time())
        store.lifecycle(iid, "resolved", {})
        return {"ok": True}

    @app.get("/incidents/{iid}/thoughts")
    async def thoughts(iid: str, store: OrgStore = Depends(auth)):
        return store.thoughts(iid)

    @app.get("/incidents/{iid}/citations")
    async def citations(iid: str, store: OrgStore = Depends(auth)):
        return store.citations(iid)

    @app.get("/incidents/{iid}/suggestions")
    async def suggestions(iid: str, store: OrgStore = Depends(auth)):
        return store.suggestions(iid)

    @app.get("/incidents/{iid}/findings")
    async def findings(iid: str, store: OrgStore = Depends(auth)):
        return store.findings(iid)

    @app.get("/incidents/{iid}/timeline")
    async def timeline(iid: str, store: OrgStore = Depends(auth)):
        return store.lifecycle_events(iid)

    # ---- SSE stream (reference: incidents_sse.py:34) ----
    @app.get("/incidents/stream")
    async def sse(store: OrgStore = Depends(auth)):
        org_id = store.org_id

        async def gen():
            q = rt.events.subscribe(org_id)
            try:
                yield "retry: 3000\n\n"
                while True:
                    try:
                        ev = await asyncio.wait_for(q.get(), timeout=15.0)
                        yield f"data: {json.dumps(ev)}\n\n"
                    except asyncio.TimeoutError:
                        yield ": keepalive\n\n"
            finally:
                rt.events.unsubscribe(org_id, q)

        return StreamingResponse(gen(), media_type="text/event-stream")

    # ---- monitor: waterfall + fleet (reference: monitor/waterfall_routes.py) ----
    @app.get("/monitor/waterfall/{iid}")
    async def waterfall(iid: str, store: OrgStore = Depends(auth)):
        if store.get_incident(iid) is None:  # no cross-org probing oracle
            raise HTTPException(404, "unknown incident")
        steps = store.steps(iid)
        return {"incident_id": iid, "steps": steps,
                "total_ms": sum(((s["finished_at"] or s["started_at"])
                                 - s["started_at"]) * 1000 for s in steps)}

    @app.get("/monitor/tool-stats")
    async def tool_stats(store: OrgStore = Depends(auth)):
        rows = rt.db.query(
            "SELECT tool, COUNT(*) n, AVG(COALESCE(finished_at,started_at)-started_at) "
            "avg_s, SUM(status='error') errors FROM execution_steps WHERE org_id=? "
            "GROUP BY tool", [store.org_id])
        return [dict(r) for r in rows]

    @app.get("/monitor/fleet")
    async def fleet(store: OrgStore = Depends(auth)):
        running = store.list_incidents()
        return {"running": [i for i in running if i["aurora_status"] == "running"],
                "queue_depth": rt.queue.pending_count(),
                "engine": rt.engine.counters()}

    # ---- metrics (reference: metrics_routes.py: MTTD / pickup latency) ----
    @app.get("/metrics")
    async def metrics(store: OrgStore = Depends(auth)):
        rows = store.list_incidents(limit=500)
        pickup = [r["investigation_started_at"] - r["created_at"]
                  for r in rows if r.get("investigation_started_at")]
        done = [r for r in rows if r["aurora_status"] == "completed"]
        ttr = [r["updated_at"] - r["created_at"] for r in done]
        sev = {}
        for r in rows:
            sev[r.get("severity") or "unknown"] = sev.get(r.get("severity") or "unknown", 0) + 1
        med = (lambda xs: sorted(xs)[len(xs) // 2] if xs else None)
        return {
            "incidents_total": len(rows),
            "incidents_completed": len(done),
            "severity_mix": sev,
            "pickup_latency_p50_s": med(pickup),
            "time_to_rca_p50_s": med(ttr),
            "engine_counters": rt.engine.counters(),
        }

    # ---- postmortems / artifacts ----
    @app.get("/postmortems/{iid}")
    async def get_pm(iid: str, store: OrgStore = Depends(auth)):
        pm = store.get_postmortem(iid)
        if pm is None:
            raise HTTPException(404, "no postmortem")
        return pm

    @app.put("/postmortems/{iid}")
    async def put_pm(iid: str, body: dict, store: OrgStore = Depends(auth),
                     x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        return {"id": store.save_postmortem(iid, body.get("content", ""))}

    @app.get("/artifacts")
    async def list_artifacts(session_id: Optional[str] = None,
                             store: OrgStore = Depends(auth)):
        return store.artifacts(session_id)

    # ---- llm usage / audit ----
    @app.get("/llm-usage")
    async def llm_usage(store: OrgStore = Depends(auth)):
        from ..utils.billing import usage_with_cost
        return usage_with_cost(store.usage_summary())

    @app.get("/audit")
    async def audit_log(store: OrgStore = Depends(auth)):
        return store.audit_events()

    # ---- command policies ----
    @app.get("/command-policies")
    async def get_policies(store: OrgStore = Depends(auth)):
        return store.command_policies()

    @app.post("/command-policies")
    async def add_policy(body: dict, store: OrgStore = Depends(auth),
                         x_user_id: Optional[str] = Header(None)):
        _require("admin", x_user_id)
        if body.get("action") not in ("allow", "deny"):
            raise HTTPException(400, "action must be allow|deny")
        return {"id": store.add_command_policy(body["pattern"], body["action"])}

    @app.delete("/command-policies/{pid}")
    async def del_policy(pid: str, store: OrgStore = Depends(auth),
                         x_user_id: Optional[str] = Header(None)):
        _require("admin", x_user_id)
        return {"deleted": store.delete_command_policy(pid)}

    # ---- knowledge base ----
    @app.post("/knowledge-base/documents")
    async def kb_upload(body: dict, store: OrgStore = Depends(auth),
                        x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        name = body.get("name", "untitled")
        content = body.get("content", "")
        did = store.add_kb_document(name, content)
        from ..rag.index import chunk_text
        for i, ch in enumerate(chunk_text(content)):
            store.add_kb_chunk(did, i, ch)
        rt.rag(store.org_id).add_document(did, name, content)
        return {"document_id": did}

    @app.get("/knowledge-base/documents")
    async def kb_list(store: OrgStore = Depends(auth)):
        return [{k: d[k] for k in ("id", "name", "ts")} for d in store.kb_documents()]

    @app.get("/knowledge-base/search")
    async def kb_search(q: str, k: int = 5, store: OrgStore = Depends(auth)):
        return rt.rag(store.org_id).search(q, k=k)

    # ---- infra graph (reference: graph_routes.py) ----
    @app.get("/graph/services")
    async def graph_services(store: OrgStore = Depends(auth)):
        return rt.graph(store.org_id).list_services()

    @app.get("/graph/impact/{service}")
    async def graph_impact(service: str, store: OrgStore = Depends(auth)):
        return rt.graph(store.org_id).impact_radius(service)

    @app.get("/graph/overview")
    async def graph_overview(store: OrgStore = Depends(auth)):
        return rt.graph(store.org_id).overview()

    @app.post("/graph/services")
    async def graph_upsert(body: dict, store: OrgStore = Depends(auth),
                           x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        g = rt.graph(store.org_id)
        g.upsert_service(body["name"], kind=body.get("kind", "service"))
        for dep in body.get("depends_on", []):
            g.upsert_service(dep)
            g.add_dependency(body["name"], dep)
        return {"ok": True}

    # ---- actions (reference: services/actions routes) ----
    @app.get("/actions")
    async def list_actions(store: OrgStore = Depends(auth)):
        return store._select("actions")

    @app.post("/actions")
    async def create_action(body: dict, store: OrgStore = Depends(auth),
                            x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        if body.get("trigger") not in ("on_incident", "schedule"):
            raise HTTPException(400, "trigger must be on_incident|schedule")
        from .store import new_id
        aid = new_id("act")
        store._insert("actions", dict(
            id=aid, name=str(body.get("name", "unnamed"))[:200],
            trigger=body["trigger"], enabled=1, created_at=time.time(),
            config=json.dumps(body.get("config") or {}), last_run=0))
        return {"id": aid}

    @app.post("/actions/{aid}/toggle")
    async def toggle_action(aid: str, store: OrgStore = Depends(auth),
                            x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        row = store._select("actions", "id=?", [aid])
        if not row:
            raise HTTPException(404, "no such action")
        new = 0 if row[0]["enabled"] else 1
        store._update("actions", {"enabled": new}, "id=?", [aid])
        return {"id": aid, "enabled": bool(new)}

    @app.delete("/actions/{aid}")
    async def delete_action(aid: str, store: OrgStore = Depends(auth),
                            x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        return {"deleted": store._delete("actions", "id=?", [aid]) > 0}

    # ---- skills catalog (reference: skills registry surface) ----
    @app.get("/skills")
    async def skills_catalog(store: OrgStore = Depends(auth)):
        from ..agent.skills.registry import get_skill_registry
        return get_skill_registry().catalog()

    @app.get("/skills/{name}")
    async def skill_body(name: str, store: OrgStore = Depends(auth)):
        from ..agent.skills.registry import get_skill_registry
        sk = get_skill_registry().get(name)
        if sk is None:
            raise HTTPException(404, "unknown skill")
        return {"name": sk.name, "description": sk.description,
                "category": sk.category, "body": sk.body}

    # ---- artifacts by name (reference: artifact_routes.py) ----
    @app.get("/artifacts/{name}")
    async def get_artifact(name: str, store: OrgStore = Depends(auth)):
        row = store.get_artifact(name)
        if row is None:
            raise HTTPException(404, "no such artifact")
        return row

    @app.put("/artifacts/{name}")
    async def put_artifact(name: str, body: dict,
                           store: OrgStore = Depends(auth),
                           x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        store.save_artifact(name, str(body.get("content", ""))[:200000],
                            session_id=body.get("session_id"))
        return {"ok": True}

    # ---- incident mutation/extras ----
    @app.patch("/incidents/{iid}")
    async def patch_incident(iid: str, body: dict,
                             store: OrgStore = Depends(auth),
                             x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        allowed = {k: v for k, v in body.items()
                   if k in ("title", "severity", "status", "user_id")}
        if not allowed:
            raise HTTPException(400, "no mutable fields given "
                                     "(title/severity/status/user_id)")
        if store.update_incident(iid, **allowed) == 0:
            raise HTTPException(404, "no such incident")
        store.lifecycle(iid, "updated", allowed)
        return {"ok": True, "updated": sorted(allowed)}

    @app.get("/incidents/{iid}/hypotheses")
    async def hypotheses(iid: str, store: OrgStore = Depends(auth)):
        row = store.get_artifact(f"hypothesis_ledger:{iid}")
        if row is None:
            return {"hypotheses": [], "ruled_out": [], "unexplored": []}
        return json.loads(row["content"])

    @app.get("/sessions")
    async def list_sessions(limit: int = 50,
                            store: OrgStore = Depends(auth)):
        rows = store._select("chat_sessions", order="created_at DESC",
                             limit=max(1, min(int(limit), 200)))
        return [{k: r[k] for k in ("id", "user_id", "is_background",
                                   "incident_id", "created_at")}
                for r in rows]

    # ---- knowledge base delete ----
    @app.delete("/knowledge-base/documents/{did}")
    async def kb_delete(did: str, store: OrgStore = Depends(auth),
                        x_user_id: Optional[str] = Header(None)):
        _require("write", x_user_id)
        store.delete_kb_document(did)
        rt._rag.pop(store.org_id, None)  # rebuild the index lazily
        return {"deleted": did}

    # ---- webhook secret management (reference: per-connector secret) ----
    @app.post("/org/webhook-secrets")
    async def set_webhook_secret(body: dict, store: OrgStore = Depends(auth),
                                 x_user_id: Optional[str] = Header(None)):
        _require("admin", x_user_id)
        source = body.get("source", "")
        if source not in sources() and source not in ("github", "bitbucket"):
            raise HTTPException(400, f"unknown source {source!r}")
        store.store_tokens("system", f"webhook_secret:{source}",
                           str(body.get("secret", "")))
        store.audit("webhook_secret_set", source)
        return {"ok": True, "source": source}

    # ---- engine status (reference: engine counters surface) ----
    @app.get("/engine/status")
    async def engine_status(store: OrgStore = Depends(auth)):
        eng = rt.engine.engine
        return {"healthy": rt.engine.healthy(),
                "model": rt.settings.main_model,
                "device": str(getattr(eng, "device", "cpu")),
                "counters": rt.engine.counters(),
                "kv_pages_free": eng.kv.num_free(),
                "kv_pages_total": eng.kv.num_pages,
                "running": eng.num_running(),
                "aux_models": sorted(rt._aux_engines)}

    @app.post("/connectors/{provider}/disconnect")
    async def disconnect(provider: str, store: OrgStore = Depends(auth),
                         x_user_id: Optional[str] = Header(None)):
        _require("connect", x_user_id)
        n = store._delete("connector_tokens", "user_id=? AND provider=?",
                          [x_user_id or "", provider])
        store.audit("connector_disconnected", f"{provider} by {x_user_id}")
        return {"disconnected": n > 0}

    # ---- connectors ----
    @app.get("/connectors")
    async def connectors(store: OrgStore = Depends(auth),
                         x_user_id: Optional[str] = Header(None)):
        from ..connectors.registry import connector_catalog
        connected = store.connected_providers(x_user_id or "")
        return [dict(c, connected=c["name"] in connected)
                for c in connector_catalog()]

    @app.get("/connectors/{provider}/oauth/start")
    async def oauth_start(provider: str, redirect_uri: str = "",
                          store: OrgStore = Depends(auth),
                          x_user_id: Optional[str] = Header(None)):
        """Begin the OAuth authorization-code flow (reference: per-connector
        OAuth routes, connectors/<name>_connector; state is one-shot CSRF)."""
        _require("connect", x_user_id)
        from ..connectors import oauth as _oauth
        from ..connectors.registry import get_connector
        spec = get_connector(provider)
        if spec is None or spec.auth != "oauth":
            raise HTTPException(404, f"{provider} is not an oauth connector")
        app_creds = store.get_token_ref("system", f"oauth_app:{provider}")
        from ..utils.secrets import secrets_backend
        cfg = secrets_backend().get(app_creds) if app_creds else None
        if not cfg or not cfg.get("client_id"):
            raise HTTPException(
                400, f"no OAuth app registered for {provider}: store "
                     f"client_id/client_secret under oauth_app:{provider}")
        state = rt.oauth_states.issue(x_user_id or "", provider)
        url = _oauth.authorize_url(provider, cfg["client_id"],
                                   redirect_uri or cfg.get("redirect_uri", ""),
                                   state)
        return {"authorize_url": url, "state": state}

    @app.get("/connectors/oauth/callback")
    async def oauth_callback(code: str, state: str,
                             store: OrgStore = Depends(auth)):
        """Code -> token exchange; credentials land in the secrets backend,
        the DB keeps only the reference (AGENTS.md:33)."""
        from ..connectors import oauth as _oauth
        from ..connectors.clients import http_transport
        from ..utils.secrets import secrets_backend
        st = rt.oauth_states.consume(state)
        if st is None:
            raise HTTPException(403, "unknown or expired OAuth state")
        provider = st["provider"]
        app_ref = store.get_token_ref("system", f"oauth_app:{provider}")
        cfg = secrets_backend().get(app_ref) if app_ref else None
        if not cfg:
            raise HTTPException(400, "OAuth app config disappeared")
        transport = getattr(rt, "provider_transport", None) or http_transport()
        try:
            creds = await _oauth.exchange_code(
                transport, provider, cfg["client_id"],
                cfg.get("client_secret", ""), code,
                cfg.get("redirect_uri", ""))
        except Exception as e:  # noqa: BLE001 - no egress / provider error
            raise HTTPException(502, f"token exchange failed: {e}") from None
        from ..connectors.registry import connect_provider
        return connect_provider(rt, store, st["user_id"], provider,
                                {"credentials": creds})

    @app.post("/connectors/{provider}/connect")
    async def connect(provider: str, body: dict, store: OrgStore = Depends(auth),
                      x_user_id: Optional[str] = Header(None)):
        _require("connect", x_user_id)
        from ..connectors.registry import connect_provider
        return connect_provider(rt, store, x_user_id or "", provider, body)

    # ---- org management (reference: org routes; before_add_member hook) ----
    @app.get("/org/users")
    async def org_users(store: OrgStore = Depends(auth)):
        rows = rt.db.query("SELECT id, email, role FROM users WHERE org_id=?",
                           [store.org_id])
        return [dict(r) for r in rows]

    @app.post("/org/users")
    async def add_user(body: dict, store: OrgStore = Depends(auth),
                       x_user_id: Optional[str] = Header(None)):
        _require("admin", x_user_id)
        from ..utils.hooks import fire
        if not await fire("before_add_member", org_id=store.org_id,
                          email=body.get("email"), default=True):
            raise HTTPException(403, "membership denied by hook")
        uid = rt.admin.create_user(store.org_id, body["email"],
                                   role=body.get("role", "member"))
        store.audit("user_added", body["email"])
        return {"user_id": uid}

    @app.post("/org/tool-permissions")
    async def set_tool_perm(body: dict, store: OrgStore = Depends(auth),
                            x_user_id: Optional[str] = Header(None)):
        _require("admin", x_user_id)
        store.set_tool_permission(body["tool"], bool(body["allowed"]))
        return {"ok": True}

    @app.get("/org/tool-permissions")
    async def get_tool_perms(store: OrgStore = Depends(auth)):
        return store.tool_permissions()

    # ---- health ----
    @app.get("/ui")
    async def ui():
        """Built-in single-file operations dashboard (the reference's
        Next.js client consumes this same REST/SSE surface; SURVEY.md
        2.10 notes the client can be kept if the API is preserved)."""
        from fastapi.responses import HTMLResponse
        from .dashboard import DASHBOARD_HTML
        return HTMLResponse(DASHBOARD_HTML)

    @app.get("/metrics/prometheus")
    async def metrics_prometheus():
        """Prometheus exposition of engine + queue + incident gauges
        (fleet-level, unauthenticated like /health; org-level SRE numbers
        stay on the authed JSON /metrics)."""
        from fastapi.responses import PlainTextResponse
        eng = rt.engine.counters() if hasattr(rt.engine, "counters") else {}
        lines = [
            "# TYPE aurora_engine_tokens_total counter",
            f"aurora_engine_tokens_total{{kind=\"prompt\"}} {eng.get('prompt_tokens', 0)}",
            f"aurora_engine_tokens_total{{kind=\"generated\"}} {eng.get('generated_tokens', 0)}",
            "# TYPE aurora_engine_steps_total counter",
            f"aurora_engine_steps_total{{kind=\"prefill\"}} {eng.get('prefill_steps', 0)}",
            f"aurora_engine_steps_total{{kind=\"decode\"}} {eng.get('decode_steps', 0)}",
            "# TYPE aurora_engine_preemptions_total counter",
            f"aurora_engine_preemptions_total {eng.get('preemptions', 0)}",
            "# TYPE aurora_prefix_cache_pages_total counter",
            f"aurora_prefix_cache_pages_total {eng.get('prefix_cache_pages', 0)}",
            "# TYPE aurora_queue_depth gauge",
            f"aurora_queue_depth {rt.queue.pending_count()}",
            "# TYPE aurora_engine_healthy gauge",
            f"aurora_engine_healthy {int(rt.engine.healthy()) if hasattr(rt.engine, 'healthy') else 1}",
        ]
        e = getattr(rt.engine, "engine", None)
        if e is not None:
            lines += [
                "# TYPE aurora_kv_pages gauge",
                f"aurora_kv_pages{{state=\"free\"}} {e.kv.num_free()}",
                f"aurora_kv_pages{{state=\"total\"}} {e.kv.num_pages}",
                "# TYPE aurora_sequences_running gauge",
                f"aurora_sequences_running {e.num_running()}",
                "# TYPE aurora_sessions_cached gauge",
                f"aurora_sessions_cached {len(e.sessions)}",
            ]
        for org in rt.admin.orgs():
            st = rt.org(org["id"])
            by = {}
            for r in st.list_incidents(limit=500):
                by[r["aurora_status"]] = by.get(r["aurora_status"], 0) + 1
            for status, n in sorted(by.items()):
                lines.append(
                    f"aurora_incidents{{org=\"{org['id']}\",status=\"{status}\"}} {n}")
        return PlainTextResponse("\n".join(lines) + "\n",
                                 media_type="text/plain; version=0.0.4")

    @app.get("/health")
    async def health():
        ok = rt.engine.healthy() if hasattr(rt.engine, "healthy") else True
        return {"status": "ok" if ok else "engine_stuck",
                "engine": rt.engine.counters(),
                "queue_depth": rt.queue.pending_count()}

    return app
