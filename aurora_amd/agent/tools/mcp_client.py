"""External MCP client: spawn stdio MCP servers, discover their tools and
expose them to the agent as gated registry tools.

Rebuilds the reference's RealMCPServerManager (reference:
chat/backend/agent/tools/mcp_tools.py:97-298): stdio JSON-RPC transport
(newline-delimited), per-server request serialization, credential
injection into the child environment, dead-process restart, tool →
registry conversion with destructive-tool classification (reference:
mcp_tools.py:38-67) mapping to the confirmation gate.

Server configuration is env-driven (AURORA_MCP_SERVERS = JSON object
name -> {command: [...], env: {...}, connector: "github", cwd: "..."}),
matching the reference's static server_configs table.
"""
from __future__ import annotations

import asyncio
import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from .registry import Tool, ToolContext, ToolRegistry

# ---------------------------------------------------------------- classify

#: verbs that create/modify/delete resources (reference: mcp_tools.py:38-55
#: — same contract: prefix match OR exact-name match => destructive)
_DESTRUCTIVE_PREFIXES = (
    "create_", "delete_", "update_", "push_", "merge_", "close_",
    "add_", "remove_", "cancel_", "rerun_", "fork_", "assign_",
    "request_", "submit_", "approve_", "dismiss_", "resolve_",
    "write_", "set_", "put_", "post_", "patch_", "terminate_", "restart_",
    "scale_", "deploy_", "rollback_", "apply_", "exec_", "run_",
)

_DESTRUCTIVE_EXACT = {
    "create_or_update_file", "push_files", "merge_pull_request",
    "update_pull_request_branch", "add_issue_comment", "delete_file",
    "cancel_workflow_run", "rerun_workflow_run", "rerun_failed_jobs",
}


def is_destructive_mcp_tool(tool_name: str) -> bool:
    """True when the tool creates, modifies or deletes resources."""
    if tool_name in _DESTRUCTIVE_EXACT:
        return True
    return any(tool_name.startswith(p) for p in _DESTRUCTIVE_PREFIXES)


def summarize_mcp_action(tool_name: str, kwargs: dict) -> str:
    """Human-readable summary shown in the confirmation prompt."""
    parts = [f"The MCP tool will {tool_name.replace('_', ' ')}"]
    for key in ("repo", "branch", "path", "title", "name", "id"):
        if key in kwargs:
            parts.append(f"{key}={kwargs[key]!r}")
    return " ".join(parts)


# ---------------------------------------------------------------- transport


class MCPError(RuntimeError):
    pass


@dataclass
class MCPServerSpec:
    name: str
    command: List[str]
    env: Dict[str, str] = field(default_factory=dict)
    cwd: Optional[str] = None
    connector: Optional[str] = None  # gate tools on this connection
    description: str = ""


class MCPStdioClient:
    """One stdio MCP server: process lifecycle + serialized JSON-RPC."""

    PROTOCOL_VERSION = "2024-11-05"

    def __init__(self, spec: MCPServerSpec, request_timeout_s: float = 30.0):
        self.spec = spec
        self.request_timeout_s = request_timeout_s
        self.proc: Optional[asyncio.subprocess.Process] = None
        self._id = 0
        # stdio is a single sequential channel: serialize requests
        # (reference uses per-server locks, mcp_tools.py:135-146)
        self._lock = asyncio.Lock()
        self.server_info: Dict[str, Any] = {}
        self.tools: List[Dict[str, Any]] = []

    @property
    def alive(self) -> bool:
        return self.proc is not None and self.proc.returncode is None

    async def start(self, extra_env: Optional[Dict[str, str]] = None) -> None:
        if self.alive:
            return
        env = dict(os.environ)
        env.update(self.spec.env)
        if extra_env:
            env.update(extra_env)
        try:
            self.proc = await asyncio.create_subprocess_exec(
                *self.spec.command,
                stdin=asyncio.subprocess.PIPE,
                stdout=asyncio.subprocess.PIPE,
                stderr=asyncio.subprocess.DEVNULL,
                env=env, cwd=self.spec.cwd)
        except (FileNotFoundError, PermissionError) as e:
            raise MCPError(f"cannot spawn MCP server '{self.spec.name}': {e}") from e
        # MCP handshake: initialize -> initialized notification -> tools/list
        init = await self._request("initialize", {
            "protocolVersion": self.PROTOCOL_VERSION,
            "capabilities": {},
            "clientInfo": {"name": "aurora-mi355x", "version": "0.2"},
        })
        self.server_info = init.get("serverInfo", {})
        await self._notify("notifications/initialized", {})
        listing = await self._request("tools/list", {})
        self.tools = listing.get("tools", [])

    async def stop(self) -> None:
        if self.proc is not None:
            try:
                self.proc.terminate()
                await asyncio.wait_for(self.proc.wait(), timeout=5.0)
            except (asyncio.TimeoutError, ProcessLookupError):
                try:
                    self.proc.kill()
                except ProcessLookupError:
                    pass
            self.proc = None

    async def call_tool(self, name: str, arguments: Dict[str, Any]) -> str:
        """tools/call; returns concatenated text content."""
        res = await self._request("tools/call",
                                  {"name": name, "arguments": arguments})
        if res.get("isError"):
            texts = [c.get("text", "") for c in res.get("content", [])]
            raise MCPError("; ".join(t for t in texts if t) or "MCP tool error")
        out = []
        for c in res.get("content", []):
            if c.get("type") == "text":
                out.append(c.get("text", ""))
            else:
                out.append(json.dumps(c)[:2000])
        return "\n".join(out)

    # ---- wire ----

    async def _send(self, msg: dict) -> None:
        assert self.proc is not None and self.proc.stdin is not None
        self.proc.stdin.write(json.dumps(msg).encode() + b"\n")
        await self.proc.stdin.drain()

    async def _request(self, method: str, params: dict) -> dict:
        if not self.alive:
            raise MCPError(f"MCP server '{self.spec.name}' is not running")
        async with self._lock:
            self._id += 1
            rid = self._id
            await self._send({"jsonrpc": "2.0", "id": rid,
                              "method": method, "params": params})
            while True:
                line = await asyncio.wait_for(self.proc.stdout.readline(),
                                              timeout=self.request_timeout_s)
                if not line:
                    raise MCPError(
                        f"MCP server '{self.spec.name}' closed its stdout")
                try:
                    msg = json.loads(line)
                except json.JSONDecodeError:
                    continue  # stray log line on stdout
                if msg.get("id") != rid:
                    continue  # notification or stale response
                if "error" in msg:
                    err = msg["error"]
                    raise MCPError(f"{err.get('message')} ({err.get('code')})")
                return msg.get("result", {})

    async def _notify(self, method: str, params: dict) -> None:
        await self._send({"jsonrpc": "2.0", "method": method, "params": params})


# ---------------------------------------------------------------- manager


def _specs_from_env() -> Dict[str, MCPServerSpec]:
    raw = os.environ.get("AURORA_MCP_SERVERS", "")
    if not raw:
        return {}
    try:
        cfg = json.loads(raw)
    except json.JSONDecodeError:
        return {}
    out = {}
    for name, c in cfg.items():
        if not isinstance(c, dict) or not c.get("command"):
            continue
        out[name] = MCPServerSpec(
            name=name, command=list(c["command"]), env=dict(c.get("env") or {}),
            cwd=c.get("cwd"), connector=c.get("connector"),
            description=c.get("description", ""))
    return out


class MCPManager:
    """Spawns configured MCP servers and registers their tools as
    ``mcp_<server>_<tool>`` registry entries (destructive ones carry the
    confirmation gate; connector-gated ones require the connection)."""

    def __init__(self, specs: Optional[Dict[str, MCPServerSpec]] = None,
                 secrets=None):
        self.specs = specs if specs is not None else _specs_from_env()
        self.clients: Dict[str, MCPStdioClient] = {}
        self.secrets = secrets  # optional: ref -> credential env injection

    async def client(self, name: str) -> MCPStdioClient:
        spec = self.specs.get(name)
        if spec is None:
            raise MCPError(f"no MCP server configured under '{name}'")
        cli = self.clients.get(name)
        if cli is None or not cli.alive:  # dead-process restart
            cli = MCPStdioClient(spec)
            extra_env = {}
            if self.secrets is not None and spec.connector:
                ref = f"{spec.connector}:mcp"
                cred = self.secrets.get(ref)
                if isinstance(cred, dict):
                    extra_env = {str(k): str(v) for k, v in cred.items()}
            await cli.start(extra_env=extra_env)
            self.clients[name] = cli
        return cli

    async def stop_all(self) -> None:
        for cli in self.clients.values():
            await cli.stop()
        self.clients.clear()

    async def register_tools(self, registry: ToolRegistry) -> List[str]:
        """Discover every configured server's tools and register them.
        Returns the registered tool names. Unreachable servers are skipped
        (their tools simply don't appear — same degradation the reference
        uses when a server binary is absent, mcp_tools.py:210-216)."""
        registered: List[str] = []
        for name in list(self.specs):
            try:
                cli = await self.client(name)
            except (MCPError, asyncio.TimeoutError):
                continue
            for t in cli.tools:
                tname = f"mcp_{name}_{t.get('name', '')}"
                destructive = is_destructive_mcp_tool(t.get("name", ""))
                params = {}
                schema = t.get("inputSchema") or {}
                for pname, p in (schema.get("properties") or {}).items():
                    params[pname] = str(p.get("description", p.get("type", "")))
                registry.register(Tool(
                    name=tname,
                    description=(t.get("description", "") or "")[:400]
                    or f"{name} MCP tool {t.get('name')}",
                    func=self._make_invoker(name, t.get("name", "")),
                    parameters=params,
                    read_only=not destructive,
                    requires_connection=self.specs[name].connector,
                    needs_confirmation=destructive,
                ))
                registered.append(tname)
        return registered

    def _make_invoker(self, server: str, tool: str):
        async def invoke(ctx: ToolContext, **kwargs) -> str:
            try:
                cli = await self.client(server)
                return await cli.call_tool(tool, kwargs)
            except (MCPError, asyncio.TimeoutError) as e:
                return f"ERROR: MCP '{server}/{tool}': {e}"
        invoke.__name__ = f"mcp_{server}_{tool}"
        return invoke
