"""MCP server: exposes Aurora to IDE clients (Cursor/Claude Desktop style).

Rebuilds the reference's inbound MCP surface (reference: mcp_server.py:50,
130,360 bearer-token middleware + aurora_mcp/registry.py:75,155,973,1098 —
always-on vs connector-gated tools, dispatch-entry catalog mapping tool
names to API routes, tool search, safety assertion).

Protocol: MCP JSON-RPC 2.0 over stdio (initialize / tools/list /
tools/call) — implemented directly, no SDK in the wheelhouse.
"""
from __future__ import annotations

import asyncio
import json
import sys
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional


@dataclass
class DispatchEntry:
    name: str
    description: str
    params: Dict[str, str]
    handler: Callable  # (rt, store, args) -> awaitable result
    gated_on: Optional[str] = None  # connector gate


# kubectl-style exec tools are deliberately NOT exposed over MCP
# (reference: aurora_mcp/registry.py:973 assert_allowlist_safe)
BANNED_OVER_MCP = {"terminal_exec", "kubectl", "cloud_exec"}


def _entries() -> List[DispatchEntry]:
    async def list_incidents(rt, store, a):
        return store.list_incidents(a.get("status"))

    async def get_incident(rt, store, a):
        return store.get_incident(a["incident_id"])

    async def investigate(rt, store, a):
        tid = rt.queue.enqueue("run_background_chat", {
            "org_id": store.org_id, "incident_id": a["incident_id"]})
        return {"task_id": tid}

    async def kb_search(rt, store, a):
        return rt.rag(store.org_id).search(a["query"], k=int(a.get("k", 5)))

    async def get_postmortem(rt, store, a):
        return store.get_postmortem(a["incident_id"])

    async def metrics(rt, store, a):
        rows = store.list_incidents(limit=200)
        return {"total": len(rows),
                "completed": sum(r["aurora_status"] == "completed" for r in rows)}

    async def graph_impact(rt, store, a):
        return rt.graph(store.org_id).impact_radius(a["service"])

    async def create_incident(rt, store, a):
        return {"incident_id": store.create_incident(
            a.get("title", "mcp incident"), "mcp", a.get("payload", {}))}

    return [
        DispatchEntry("aurora_list_incidents", "List recent incidents",
                      {"status": "optional status filter"}, list_incidents),
        DispatchEntry("aurora_get_incident", "Get one incident with RCA summary",
                      {"incident_id": "incident id"}, get_incident),
        DispatchEntry("aurora_investigate", "Dispatch a background RCA",
                      {"incident_id": "incident id"}, investigate),
        DispatchEntry("aurora_create_incident", "Create an incident",
                      {"title": "title", "payload": "alert payload"}, create_incident),
        DispatchEntry("aurora_search_knowledge_base", "Hybrid KB search",
                      {"query": "query", "k": "results"}, kb_search),
        DispatchEntry("aurora_get_postmortem", "Fetch a postmortem",
                      {"incident_id": "incident id"}, get_postmortem),
        DispatchEntry("aurora_metrics", "Incident metrics summary", {}, metrics),
        DispatchEntry("aurora_graph_impact", "Service impact radius",
                      {"service": "service name"}, graph_impact),
    ]


def search_dispatch_entries(query: str) -> List[DispatchEntry]:
    q = query.lower()
    return [e for e in _entries()
            if q in e.name.lower() or q in e.description.lower()]


class MCPServer:
    def __init__(self, rt, resolve_token: Optional[Callable] = None):
        self.rt = rt
        self.entries = {e.name: e for e in _entries()}
        assert not (set(self.entries) & BANNED_OVER_MCP)
        self._resolve = resolve_token or self._default_resolve

    def _default_resolve(self, token: str) -> Optional[dict]:
        """Bearer token '<user_id>' resolved against the user table
        (reference: mcp_server.py:130 _resolve_token)."""
        return self.rt.admin.get_user(token)

    async def handle(self, msg: dict, token: str) -> Optional[dict]:
        if not isinstance(msg, dict):
            return self._err(None, -32600, "invalid request")
        mid = msg.get("id")
        method = msg.get("method", "")
        if not isinstance(method, str):
            return self._err(mid, -32600, "method must be a string")
        if method == "initialize":
            return self._ok(mid, {
                "protocolVersion": "2024-11-05",
                "serverInfo": {"name": "aurora-mi355x", "version": "0.1.0"},
                "capabilities": {"tools": {}}})
        if method == "notifications/initialized":
            return None
        user = self._resolve(token)
        if user is None:
            return self._err(mid, -32001, "unauthorized")
        store = self.rt.org(user["org_id"])
        if method == "tools/list":
            return self._ok(mid, {"tools": [
                {"name": e.name, "description": e.description,
                 "inputSchema": {"type": "object", "properties": {
                     k: {"type": "string", "description": v}
                     for k, v in e.params.items()}}}
                for e in self.entries.values()]})
        if method == "tools/call":
            p = msg.get("params", {})
            if not isinstance(p, dict):
                return self._err(mid, -32602, "params must be an object")
            entry = self.entries.get(p.get("name", ""))
            if entry is None:
                return self._err(mid, -32602, f"unknown tool {p.get('name')}")
            args = p.get("arguments") or {}
            if not isinstance(args, dict):
                return self._err(mid, -32602, "arguments must be an object")
            try:
                result = await entry.handler(self.rt, store, args)
                return self._ok(mid, {"content": [
                    {"type": "text", "text": json.dumps(result, default=str)[:20000]}]})
            except Exception as e:  # noqa: BLE001
                return self._err(mid, -32000, str(e)[:500])
        return self._err(mid, -32601, f"unknown method {method}")

    @staticmethod
    def _ok(mid, result) -> dict:
        return {"jsonrpc": "2.0", "id": mid, "result": result}

    @staticmethod
    def _err(mid, code, message) -> dict:
        return {"jsonrpc": "2.0", "id": mid,
                "error": {"code": code, "message": message}}

    async def serve_stdio(self, token: str) -> None:  # pragma: no cover
        loop = asyncio.get_running_loop()
        while True:
            line = await loop.run_in_executor(None, sys.stdin.readline)
            if not line:
                return
            try:
                msg = json.loads(line)
            except json.JSONDecodeError:
                continue
            resp = await self.handle(msg, token)
            if resp is not None:
                sys.stdout.write(json.dumps(resp) + "\n")
                sys.stdout.flush()
