"""Python client SDK for the aurora-mi355x API + chat WebSocket.

The reference ships a Next.js client that proxies every backend call
(reference: client/ 78k LoC, AGENTS.md:94 forwardRequest); programmatic
users get this SDK: REST surface + the chat WS protocol (init/query/
confirmation_response) over the same hand-rolled frame layer the server
uses."""
from __future__ import annotations

import asyncio
import json
from typing import AsyncIterator, List, Optional

import httpx

from .service.ws import mask_frame, read_frame


class AuroraClient:
    def __init__(self, base_url: str = "http://127.0.0.1:5080",
                 user_id: str = "", org_id: str = "",
                 internal_secret: str = "", auth_token: str = "",
                 signing_secret: str = ""):
        """``signing_secret`` (= the server's INTERNAL_API_SECRET) lets a
        trusted caller mint its own HMAC identity token for ``user_id``;
        ``auth_token`` supplies a pre-issued one (POST /auth/token)."""
        headers = {}
        if user_id:
            headers["X-User-Id"] = user_id
        if org_id:
            headers["X-Org-Id"] = org_id
        if internal_secret:
            headers["X-Internal-Api-Secret"] = internal_secret
        if not auth_token and signing_secret and user_id:
            from .utils.auth_tokens import mint_token
            auth_token = mint_token(user_id, signing_secret, ttl_s=3600)
        if auth_token:
            headers["Authorization"] = f"Bearer {auth_token}"
        self.http = httpx.Client(base_url=base_url, headers=headers, timeout=30.0)
        self.user_id = user_id

    def _get(self, path: str):
        r = self.http.get(path)
        r.raise_for_status()
        return r.json()

    def _post(self, path: str, payload: dict):
        r = self.http.post(path, json=payload)
        r.raise_for_status()
        return r.json()

    # ---- incidents ----
    def incidents(self, status: Optional[str] = None) -> List[dict]:
        return self.http.get("/incidents", params={"status": status}
                             if status else None).json()

    def incident(self, iid: str) -> dict:
        return self.http.get(f"/incidents/{iid}").json()

    def create_incident(self, title: str, payload: Optional[dict] = None) -> str:
        return self.http.post("/incidents", json={
            "title": title, "payload": payload or {}}).json()["incident_id"]

    def investigate(self, iid: str) -> Optional[str]:
        return self.http.post(f"/incidents/{iid}/investigate").json().get("task_id")

    def send_webhook(self, source: str, org_id: str, payload: dict) -> dict:
        return self.http.post(f"/webhook/{source}/{org_id}", json=payload).json()

    def postmortem(self, iid: str) -> Optional[str]:
        r = self.http.get(f"/postmortems/{iid}")
        return r.json().get("content") if r.status_code == 200 else None

    def metrics(self) -> dict:
        return self.http.get("/metrics").json()

    def findings(self, iid: str) -> List[dict]:
        return self._get(f"/incidents/{iid}/findings")

    def citations(self, iid: str) -> List[dict]:
        return self._get(f"/incidents/{iid}/citations")

    def suggestions(self, iid: str) -> List[dict]:
        return self._get(f"/incidents/{iid}/suggestions")

    def timeline(self, iid: str) -> List[dict]:
        return self._get(f"/incidents/{iid}/timeline")

    def thoughts(self, iid: str) -> List[dict]:
        return self._get(f"/incidents/{iid}/thoughts")

    def cancel_investigation(self, iid: str) -> dict:
        return self._post(f"/incidents/{iid}/cancel", {})

    def resolve(self, iid: str) -> dict:
        return self._post(f"/incidents/{iid}/resolve", {})

    def graph_overview(self) -> dict:
        return self._get("/graph/overview")

    def impact(self, service: str) -> dict:
        return self._get(f"/graph/impact/{service}")

    def llm_usage(self) -> List[dict]:
        return self._get("/llm-usage")

    def waterfall(self, iid: str) -> dict:
        return self.http.get(f"/monitor/waterfall/{iid}").json()

    def kb_upload(self, name: str, content: str) -> str:
        return self.http.post("/knowledge-base/documents", json={
            "name": name, "content": content}).json()["document_id"]

    def kb_search(self, query: str, k: int = 5) -> List[dict]:
        return self.http.get("/knowledge-base/search",
                             params={"q": query, "k": k}).json()

    # ---- round-2 surface ----
    def actions(self) -> List[dict]:
        return self._get("/actions")

    def create_action(self, name: str, trigger: str, config: dict) -> str:
        return self._post("/actions", {"name": name, "trigger": trigger,
                                       "config": config})["id"]

    def toggle_action(self, aid: str) -> dict:
        return self._post(f"/actions/{aid}/toggle", {})

    def delete_action(self, aid: str) -> dict:
        r = self.http.delete(f"/actions/{aid}")
        r.raise_for_status()
        return r.json()

    def skills(self) -> List[dict]:
        return self._get("/skills")

    def skill(self, name: str) -> dict:
        return self._get(f"/skills/{name}")

    def artifact(self, name: str) -> dict:
        return self._get(f"/artifacts/{name}")

    def save_artifact(self, name: str, content: str) -> dict:
        r = self.http.put(f"/artifacts/{name}", json={"content": content})
        r.raise_for_status()
        return r.json()

    def update_incident(self, iid: str, **fields) -> dict:
        r = self.http.patch(f"/incidents/{iid}", json=fields)
        r.raise_for_status()
        return r.json()

    def hypotheses(self, iid: str) -> dict:
        return self._get(f"/incidents/{iid}/hypotheses")

    def sessions(self, limit: int = 50) -> List[dict]:
        return self._get(f"/sessions?limit={limit}")

    def engine_status(self) -> dict:
        return self._get("/engine/status")

    def oauth_start(self, provider: str, redirect_uri: str = "") -> dict:
        return self._get(f"/connectors/{provider}/oauth/start"
                         + (f"?redirect_uri={redirect_uri}"
                            if redirect_uri else ""))

    def disconnect_provider(self, provider: str) -> dict:
        return self._post(f"/connectors/{provider}/disconnect", {})

    def connect_provider(self, provider: str, credentials: dict) -> dict:
        return self.http.post(f"/connectors/{provider}/connect",
                              json={"credentials": credentials}).json()


class AuroraChat:
    """Async chat client over the WS protocol."""

    def __init__(self, host: str, port: int, token: str):
        self.host, self.port, self.token = host, port, token
        self.reader: Optional[asyncio.StreamReader] = None
        self.writer: Optional[asyncio.StreamWriter] = None
        self.session_id: Optional[str] = None

    async def connect(self, mode: str = "agent", confirm_tools: bool = False,
                      incident_id: Optional[str] = None) -> None:
        self.reader, self.writer = await asyncio.open_connection(self.host, self.port)
        self.writer.write((
            f"GET /?token={self.token} HTTP/1.1\r\nHost: {self.host}\r\n"
            f"Upgrade: websocket\r\nConnection: Upgrade\r\n"
            f"Sec-WebSocket-Key: Y2xpZW50LXNkay1rZXk=\r\n"
            f"Sec-WebSocket-Version: 13\r\n\r\n").encode())
        await self.writer.drain()
        resp = await self.reader.readuntil(b"\r\n\r\n")
        if b"101" not in resp.split(b"\r\n")[0]:
            raise ConnectionError("websocket handshake rejected")
        ev = await self.recv()
        assert ev["type"] == "ready", ev
        await self.send({"type": "init", "mode": mode,
                         "confirm_tools": confirm_tools,
                         "incident_id": incident_id})
        ev = await self.recv()
        self.session_id = ev.get("session_id")

    async def send(self, obj: dict) -> None:
        self.writer.write(mask_frame(json.dumps(obj).encode()))
        await self.writer.drain()

    async def recv(self) -> dict:
        opcode, data = await read_frame(self.reader)
        if opcode == 0x8:
            raise ConnectionError("closed")
        return json.loads(data)

    async def query(self, text: str,
                    auto_approve: bool = False) -> AsyncIterator[dict]:
        """Stream events for one query until final/error; optionally
        auto-approve tool confirmation requests."""
        await self.send({"type": "query", "content": text})
        while True:
            ev = await self.recv()
            if ev["type"] == "confirmation_request" and auto_approve:
                await self.send({"type": "confirmation_response",
                                 "id": ev["id"], "approved": True})
                continue
            yield ev
            if ev["type"] in ("final", "error", "blocked"):
                return

    async def cancel(self) -> None:
        await self.send({"type": "control:cancel"})

    async def close(self) -> None:
        if self.writer:
            self.writer.close()
