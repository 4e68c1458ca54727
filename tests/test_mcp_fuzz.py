"""MCP JSON-RPC robustness fuzz: arbitrary message shapes through
MCPServer.handle must return a JSON-RPC response (or None for
notifications) and never raise (service/mcp_server.py)."""
import asyncio

import pytest
from hypothesis import given, settings, strategies as st

from aurora_amd.engine.engine import EngineConfig
from aurora_amd.engine.server import AsyncEngineServer
from aurora_amd.service.mcp_server import MCPServer
from aurora_amd.service.runtime import Runtime
from aurora_amd.utils.config import Settings


@pytest.fixture(scope="module")
def mcp():
    server = AsyncEngineServer(EngineConfig(model="tiny-llama", device="cpu"))
    rt = Runtime(Settings(db_path=":memory:",
                          secrets_path="/tmp/mcp_fz.json"), engine_server=server)
    org = rt.admin.create_org("m")
    rt.admin.create_user(org, "m@m.io", user_id="mcp-user")
    yield MCPServer(rt)
    server.shutdown()


soup = st.recursive(
    st.none() | st.booleans() | st.integers() | st.text(max_size=20),
    lambda c: st.lists(c, max_size=3) | st.dictionaries(st.text(max_size=10), c, max_size=3),
    max_leaves=6)

messages = st.fixed_dictionaries({}, optional={
    "jsonrpc": soup, "id": soup, "method": soup, "params": soup})


@settings(max_examples=120, deadline=None)
@given(messages, st.sampled_from(["mcp-user", "not-a-user", ""]))
def test_mcp_handle_never_raises(mcp, msg, token):
    out = asyncio.run(mcp.handle(msg, token))
    assert out is None or (isinstance(out, dict) and out.get("jsonrpc") == "2.0")


def test_mcp_tool_call_bad_arguments(mcp):
    out = asyncio.run(mcp.handle(
        {"id": 1, "method": "tools/call",
         "params": {"name": "list_incidents", "arguments": "not-a-dict"}},
        "mcp-user"))
    assert "error" in out or "result" in out
    out2 = asyncio.run(mcp.handle(
        {"id": 2, "method": "tools/call", "params": ["not", "a", "dict"]},
        "mcp-user"))
    assert "error" in out2
