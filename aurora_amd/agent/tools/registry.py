"""Tool registry: assembly, gating, execution capture.

Mirrors the reference's tool layer contract (reference:
tools/cloud_tools.py:1008-1436): per-context assembly, mode gating (ask =
read-only), org tool permissions, connection-gated registration, execution
capture into execution_steps (the waterfall), output redaction + capping,
and tool_start/tool_end frames pushed to the chat transport.
"""
from __future__ import annotations

import asyncio
import json
import time
from dataclasses import dataclass, field
from typing import Any, Awaitable, Callable, Dict, List, Optional

from ...guardrails.redaction import redact

MAX_TOOL_OUTPUT_CHARS = 12000  # head+tail cap (reference: tool_output_cap.py:22)


@dataclass
class ToolContext:
    org_store: Any = None
    user_id: Optional[str] = None
    session_id: Optional[str] = None
    incident_id: Optional[str] = None
    llm: Any = None
    mode: str = "agent"
    rag: Any = None
    graph: Any = None
    queue: Any = None
    notifier: Optional[Callable[[dict], Awaitable[None]]] = None  # WS frames
    confirmer: Optional[Callable[[str, dict], Awaitable[bool]]] = None
    settings: Any = None
    depth: int = 0  # nested-RCA guard (reference: trigger_rca_tool.py:38)
    extra: Dict[str, Any] = field(default_factory=dict)


@dataclass
class Tool:
    name: str
    description: str
    func: Callable[..., Awaitable[str]]
    parameters: Dict[str, str] = field(default_factory=dict)  # name -> description
    read_only: bool = True
    requires_connection: Optional[str] = None
    needs_confirmation: bool = False  # interactive human-in-the-loop gate

    def spec_line(self) -> str:
        ps = ", ".join(f"{k}: {v}" for k, v in self.parameters.items())
        return f"- {self.name}({ps}): {self.description}"


def cap_output(text: str, limit: int = MAX_TOOL_OUTPUT_CHARS) -> str:
    if len(text) <= limit:
        return text
    head = text[: limit * 2 // 3]
    tail = text[-limit // 3:]
    return f"{head}\n...[output truncated: {len(text)} chars total]...\n{tail}"


class ToolRegistry:
    def __init__(self):
        self._tools: Dict[str, Tool] = {}

    def register(self, tool: Tool) -> Tool:
        self._tools[tool.name] = tool
        return tool

    def add(self, name: str, description: str, parameters: Optional[Dict[str, str]] = None,
            read_only: bool = True, requires_connection: Optional[str] = None,
            needs_confirmation: bool = False):
        def deco(fn):
            self.register(Tool(name, description, fn, parameters or {},
                               read_only, requires_connection, needs_confirmation))
            return fn
        return deco

    def names(self) -> List[str]:
        return sorted(self._tools)

    def get(self, name: str) -> Optional[Tool]:
        return self._tools.get(name)

    def available(self, ctx: ToolContext,
                  subset: Optional[List[str]] = None) -> List[Tool]:
        """Assemble the tool set for this context (mode/permission/connection
        gating — reference: cloud_tools.py:1337-1436, access/)."""
        perms = {}
        connected: List[str] = []
        if ctx.org_store is not None:
            perms = ctx.org_store.tool_permissions()
            if ctx.user_id:
                connected = ctx.org_store.connected_providers(ctx.user_id)
        out = []
        for t in self._tools.values():
            if subset is not None and t.name not in subset:
                continue
            if ctx.mode == "ask" and not t.read_only:
                continue
            if perms.get(t.name) is False:
                continue
            if t.requires_connection and t.requires_connection not in connected:
                continue
            out.append(t)
        return sorted(out, key=lambda t: t.name)

    async def run(self, ctx: ToolContext, name: str, args: Dict[str, Any],
                  timeout_s: float = 120.0) -> str:
        """Execute with waterfall capture, redaction, capping, WS frames
        (reference: cloud_tools.py:1075-1309 wrap_func_with_capture)."""
        tool = self._tools.get(name)
        if tool is None:
            return f"ERROR: unknown tool '{name}'. Available: {', '.join(self.names())}"
        if ctx.mode == "ask" and not tool.read_only:
            return f"ERROR: tool '{name}' is not available in ask (read-only) mode"
        # human-in-the-loop confirmation for destructive-classified tools in
        # interactive sessions (reference: confirmation_response WS type,
        # main_chatbot.py:1000-1619; destructive classification
        # mcp_tools.py:57); headless runs have no confirmer and rely on the
        # guardrail stack alone
        if tool.needs_confirmation and ctx.confirmer is not None:
            try:
                approved = await asyncio.wait_for(ctx.confirmer(name, args),
                                                  timeout=300.0)
            except asyncio.TimeoutError:
                approved = False
            if not approved:
                if ctx.org_store is not None:
                    ctx.org_store.audit("tool_declined", name)
                return f"CANCELLED: user declined '{name}'"
        step_id = None
        if ctx.org_store is not None:
            step_id = ctx.org_store.step_start(
                name, json.dumps(args)[:2000], ctx.incident_id, ctx.session_id)
        if ctx.notifier is not None:
            await ctx.notifier({"type": "tool_start", "tool": name, "args": args})
        t0 = time.monotonic()
        try:
            result = await asyncio.wait_for(tool.func(ctx, **args), timeout=timeout_s)
            result = cap_output(redact(str(result)))
            if step_id:
                ctx.org_store.step_end(step_id, "completed", result[:4000])
            if ctx.notifier is not None:
                await ctx.notifier({"type": "tool_end", "tool": name, "status": "ok",
                                    "elapsed_ms": (time.monotonic() - t0) * 1e3})
            return result
        except asyncio.TimeoutError:
            if step_id:
                ctx.org_store.step_end(step_id, "timeout", error="timeout")
            if ctx.notifier is not None:
                await ctx.notifier({"type": "tool_end", "tool": name, "status": "timeout"})
            return f"ERROR: tool '{name}' timed out after {timeout_s}s"
        except TypeError as e:
            if step_id:
                ctx.org_store.step_end(step_id, "error", error=str(e))
            return (f"ERROR: bad arguments for '{name}': {e}. "
                    f"Expected parameters: {list(tool.parameters)}")
        except Exception as e:  # noqa: BLE001 - tool errors feed back to the model
            if step_id:
                ctx.org_store.step_end(step_id, "error", error=str(e)[:2000])
            if ctx.notifier is not None:
                await ctx.notifier({"type": "tool_end", "tool": name, "status": "error"})
            return f"ERROR: tool '{name}' failed: {e}"


GLOBAL_REGISTRY = ToolRegistry()
