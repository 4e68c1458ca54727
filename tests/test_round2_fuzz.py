"""Property-based fuzzing of round-2 surfaces: ReAct turn grammar,
provider-client parsers, OAuth token parsing, MCP client resilience
(matching the reference's security-heavy test strategy)."""
import asyncio
import json

import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from aurora_amd.connectors.clients.base import ProviderAPIError

SET = settings(max_examples=60, deadline=None)


# ------------------------------------------------------ ReAct turn grammar

@SET
@given(st.binary(min_size=0, max_size=200))
def test_react_grammar_thought_accepts_any_bytes(data):
    from aurora_amd.engine.grammar import ReActTurnGrammar
    g = ReActTurnGrammar(["kubectl", "final_answer"], thought_bytes=10**6)
    # thought phase is unconstrained UNLESS the transition literal fully
    # matched inside the data — then the tool FSM starts judging
    ok = g.advance(data)
    if b"\nAction:" not in data:
        assert ok
        assert not g.is_complete()


@SET
@given(st.text(alphabet=st.characters(min_codepoint=32, max_codepoint=126),
               min_size=0, max_size=60),
       st.sampled_from(["kubectl", "final_answer", "get_incident"]),
       st.dictionaries(st.text(alphabet="abcdef", min_size=1, max_size=5),
                       st.one_of(st.integers(-10, 10),
                                 st.text(alphabet="xyz ", max_size=8)),
                       max_size=3))
def test_react_grammar_accepts_all_wellformed_turns(thought, tool, args):
    from aurora_amd.engine.grammar import ReActTurnGrammar
    g = ReActTurnGrammar(["kubectl", "final_answer", "get_incident"],
                         thought_bytes=10**6)
    body = json.dumps({"tool": tool, "args": args},
                      separators=(",", ":"))
    s = (thought + "\nAction: " + body).encode()
    assert g.advance(s), s
    assert g.is_complete() == True  # noqa: E712
    assert g.tool_name() == tool


@SET
@given(st.lists(st.sampled_from(list(range(32, 127)) + [10, 13]),
                min_size=1, max_size=120))
def test_grammar_allowed_bytes_are_actually_legal(byte_list):
    """allowed_next_bytes must never offer a byte that advance() rejects
    (the sampler relies on this)."""
    from aurora_amd.engine.grammar import ReActTurnGrammar
    g = ReActTurnGrammar(["kubectl", "final_answer"], thought_bytes=20,
                         tool_soft_close=30)
    for b in byte_list:
        allowed = g.allowed_next_bytes()
        if allowed is not None and allowed != b"":
            for a in allowed:
                f = g.fork()
                assert f._advance_byte(a), (bytes([a]), allowed)
        if not g._advance_byte(b):
            break
        g.nbytes += 1


# --------------------------------------------------- provider parser fuzz

_JSONISH = st.recursive(
    st.one_of(st.none(), st.booleans(), st.integers(-5, 5),
              st.floats(allow_nan=False, allow_infinity=False),
              st.text(max_size=8)),
    lambda c: st.one_of(st.lists(c, max_size=3),
                        st.dictionaries(st.text(max_size=6), c, max_size=3)),
    max_leaves=8)


@SET
@given(st.dictionaries(st.sampled_from(
    ["data", "meta", "series", "monitors", "events", "results", "issues",
     "problems", "result", "messages", "channels", "total", "entry",
     "value", "incidents", "tests", "labels", "status", "errors"]),
    _JSONISH, max_size=4))
def test_parsers_fail_closed_on_malformed_docs(doc):
    """Every response parser either returns a string/tuple or raises
    ProviderAPIError — never an unhandled TypeError/KeyError (hostile or
    half-changed provider responses must not crash the agent turn)."""
    from aurora_amd.connectors.clients import (comms, datadog, observability,
                                               observability2)
    parsers = [
        datadog.parse_logs_response, datadog.parse_metrics_response,
        datadog.parse_monitors_response, datadog.parse_events_response,
        observability.parse_nrql_response, observability.parse_splunk_results,
        observability.parse_dynatrace_problems,
        observability.parse_dynatrace_metrics,
        comms.parse_jira_search, comms.parse_confluence_search,
        comms.parse_notion_search,
        observability2.parse_coroot_overview, observability2.parse_te_results,
        observability2.parse_netdata_data, observability2.parse_cf_analytics,
        observability2.parse_opsgenie_alerts,
        observability2.parse_incidentio_list,
        observability2.parse_flyio_metrics,
    ]
    for p in parsers:
        try:
            out = p(doc)
            assert isinstance(out, (str, tuple, list))
        except ProviderAPIError:
            pass
        except (AttributeError, TypeError, KeyError, ValueError,
                IndexError) as e:
            raise AssertionError(f"{p.__name__} leaked {type(e).__name__}: "
                                 f"{e} on {doc!r}") from e


@SET
@given(_JSONISH)
def test_oauth_token_parse_never_crashes(doc):
    from aurora_amd.connectors.oauth import parse_token_response
    try:
        out = parse_token_response(doc if isinstance(doc, dict) else {"x": doc})
        assert out["access_token"]
    except ProviderAPIError:
        pass


# ----------------------------------------------------- MCP client hostile

def test_mcp_client_survives_garbage_and_death():
    import os
    import sys
    from aurora_amd.agent.tools.mcp_client import (MCPError, MCPServerSpec,
                                                   MCPStdioClient)

    async def main():
        # a "server" that answers initialize then emits garbage and exits
        script = (
            "import sys,json\n"
            "line=sys.stdin.readline()\n"
            "m=json.loads(line)\n"
            "print(json.dumps({'jsonrpc':'2.0','id':m['id'],'result':"
            "{'serverInfo':{'name':'evil'},'capabilities':{}}}))\n"
            "sys.stdout.flush()\n"
            "sys.stdin.readline()\n"  # initialized notification
            "sys.stdin.readline()\n"  # tools/list request
            "print('not json at all')\n"
            "print('{\"jsonrpc\":\"2.0\",\"id\":999,\"result\":{}}')\n"
            "sys.stdout.flush()\n"
        )
        cli = MCPStdioClient(MCPServerSpec(
            name="evil", command=[sys.executable, "-c", script]),
            request_timeout_s=5.0)
        with pytest.raises(MCPError):
            await cli.start()  # tools/list never gets a matching id
        await cli.stop()
        # dead before handshake
        cli2 = MCPStdioClient(MCPServerSpec(
            name="dead", command=[sys.executable, "-c", "pass"]),
            request_timeout_s=3.0)
        with pytest.raises(MCPError):
            await cli2.start()
        await cli2.stop()

    asyncio.run(main())
