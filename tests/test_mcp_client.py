"""External MCP stdio client (reference: tools/mcp_tools.py:97-298):
spawn, handshake, tool discovery/conversion, destructive classification,
credential env injection, serialization, dead-server behavior."""
import asyncio
import os
import sys

import pytest

from aurora_amd.agent.tools.mcp_client import (MCPManager, MCPServerSpec,
                                               MCPStdioClient,
                                               is_destructive_mcp_tool)
from aurora_amd.agent.tools.registry import ToolContext, ToolRegistry

FAKE = [sys.executable, os.path.join(os.path.dirname(__file__),
                                     "fake_mcp_server.py")]


def spec(**kw):
    return MCPServerSpec(name="fake", command=FAKE, **kw)


def test_destructive_classification():
    for name in ("create_issue", "delete_file", "merge_pull_request",
                 "push_files", "update_thing", "apply_manifest",
                 "run_workflow"):
        assert is_destructive_mcp_tool(name), name
    for name in ("get_weather", "list_issues", "search_code", "read_file"):
        assert not is_destructive_mcp_tool(name), name


def test_handshake_list_and_call():
    async def main():
        cli = MCPStdioClient(spec())
        await cli.start()
        assert cli.server_info.get("name") == "fake-mcp"
        names = {t["name"] for t in cli.tools}
        assert names == {"get_weather", "delete_everything"}
        out = await cli.call_tool("get_weather", {"city": "berlin"})
        assert "sunny in berlin" in out
        # concurrent calls serialize over one stdio pipe without mixing
        outs = await asyncio.gather(*[
            cli.call_tool("get_weather", {"city": f"c{i}"}) for i in range(5)])
        assert sorted(outs) == sorted(f"sunny in c{i}" for i in range(5))
        await cli.stop()

    asyncio.run(main())


def test_manager_registers_gated_tools():
    async def main():
        reg = ToolRegistry()
        mgr = MCPManager(specs={"fake": spec(connector=None)})
        names = await mgr.register_tools(reg)
        assert "mcp_fake_get_weather" in names
        t = reg.get("mcp_fake_get_weather")
        assert t.read_only and not t.needs_confirmation
        assert "city" in t.parameters
        td = reg.get("mcp_fake_delete_everything")
        assert (not td.read_only) and td.needs_confirmation
        # run through the registry (capture path)
        ctx = ToolContext()
        out = await reg.run(ctx, "mcp_fake_get_weather", {"city": "oslo"})
        assert "sunny in oslo" in out
        # destructive tool is refused in ask (read-only) mode
        out = await reg.run(ToolContext(mode="ask"),
                            "mcp_fake_delete_everything", {})
        assert "not available in ask" in out
        # destructive tool passes through the confirmation gate

        async def deny(name, args):
            return False

        out = await reg.run(ToolContext(confirmer=deny),
                            "mcp_fake_delete_everything", {})
        assert out.startswith("CANCELLED")
        await mgr.stop_all()

    asyncio.run(main())


def test_credential_env_injection_and_restart():
    async def main():
        class Secrets:
            def get(self, ref):
                assert ref == "github:mcp"
                return {"FAKE_MCP_TOKEN": "tok-123"}

        mgr = MCPManager(specs={"fake": spec(connector="github")},
                         secrets=Secrets())
        cli = await mgr.client("fake")
        out = await cli.call_tool("get_weather", {"city": "x"})
        assert "token=tok-123" in out
        # kill the server; the manager restarts it on next use
        cli.proc.kill()
        await cli.proc.wait()
        cli2 = await mgr.client("fake")
        assert cli2 is not cli and cli2.alive
        out = await cli2.call_tool("get_weather", {"city": "y"})
        assert "sunny in y" in out
        await mgr.stop_all()

    asyncio.run(main())


def test_env_config_parsing(monkeypatch):
    monkeypatch.setenv("AURORA_MCP_SERVERS", '{"gh": {"command": ["x"], '
                       '"connector": "github"}, "bad": {}}')
    mgr = MCPManager()
    assert set(mgr.specs) == {"gh"}
    assert mgr.specs["gh"].connector == "github"
    monkeypatch.setenv("AURORA_MCP_SERVERS", "not json")
    assert MCPManager().specs == {}


def test_unreachable_server_skipped():
    async def main():
        reg = ToolRegistry()
        mgr = MCPManager(specs={"ghost": MCPServerSpec(
            name="ghost", command=["/nonexistent/mcp-server"])})
        names = await mgr.register_tools(reg)
        assert names == []

    asyncio.run(main())
