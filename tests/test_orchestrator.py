"""Orchestrator (multi-agent) + MCP server tests with a scripted LLM."""
import asyncio
import json

import pytest

from aurora_amd.service.store import AdminStore, Database


class FakeResult:
    def __init__(self, text):
        self.text = text
        self.token_ids = [0]
        self.prompt_tokens = 1
        self.elapsed_ms = 1.0
        self.finish_reason = "eos"


class ScriptedServer:
    """Returns scripted grammar responses; plain text elsewhere."""

    def __init__(self, script):
        self.script = list(script)

    def tokenizer(self):
        from aurora_amd.engine.tokenizer import ByteTokenizer
        return ByteTokenizer()

    def counters(self):
        return {}

    async def stream(self, ids, params, session_id=None, request_id=None):
        from aurora_amd.engine.engine import StepOutput
        g = params.grammar
        if g is not None and hasattr(g, "thought_bytes") and self.script:
            # fused ReAct turn: thought + scripted action JSON
            yield StepOutput("x", [1],
                             "thinking...\nAction: " + self.script.pop(0),
                             True, "grammar")
            return
        yield StepOutput("x", [1], "thinking...", True, "stop")

    async def complete_text(self, prompt, params, session_id=None):
        if params.grammar is not None and self.script:
            return FakeResult(self.script.pop(0))
        return FakeResult("summary text")


@pytest.fixture()
def env():
    db = Database(":memory:")
    admin = AdminStore(db)
    oid = admin.create_org("t")
    store = admin.org_store(oid)
    iid = store.create_incident("checkout 5xx", "pagerduty", {"service": "checkout"})
    return store, iid


def _mk(store, iid, script):
    from aurora_amd.agent.llm import LLMManager
    from aurora_amd.agent.tools.registry import ToolContext
    import aurora_amd.agent.tools.core  # noqa: F401
    server = ScriptedServer(script)
    llm = LLMManager(server, store=store)
    ctx = ToolContext(org_store=store, incident_id=iid, llm=llm)
    return llm, ctx


def test_orchestrator_multi_agent_flow(env):
    store, iid = env
    from aurora_amd.agent.orchestrator import Orchestrator
    from aurora_amd.agent.state import State
    script = [
        # triage: pick two roles
        json.dumps({"mode": "multi_agent",
                    "roles": ["runbook_lookup", "bogus_role",
                              "recent_change_investigator"], "waves": 1}),
        # each sub-agent immediately finishes (grammar action calls)
        json.dumps({"tool": "final_answer", "args": {"answer": "runbook says X"}}),
        json.dumps({"tool": "final_answer", "args": {"answer": "deploy at 14:00"}}),
        # synthesis: end with final RCA
        json.dumps({"decision": "end",
                    "rca": "Root cause: the 14:00 deploy broke checkout."}),
    ]
    llm, ctx = _mk(store, iid, script)
    orch = Orchestrator(llm, ctx, sub_agent_turns=2)
    state = State(question="checkout 5xx alert", incident_id=iid)

    res = asyncio.run(orch.run(state))
    assert "Root cause" in res.final
    # unknown role filtered (dispatcher._filter_known_roles analog)
    assert state.triage_decision["roles"][1] == "bogus_role"
    roles_run = {f["role"] for f in res.findings}
    assert roles_run == {"runbook_lookup", "recent_change_investigator"}
    # findings persisted with status transitions
    rows = store.findings(iid)
    assert len(rows) == 2 and all(r["status"] == "completed" for r in rows)


def test_orchestrator_subagent_failure_isolated(env):
    """One sub-agent raising must not sink the wave: its finding is
    marked failed, the other sub-agent completes, synthesis still runs
    (reference: dispatcher pre-emitted findings + failure capture)."""
    store, iid = env
    from aurora_amd.agent.orchestrator import Orchestrator
    from aurora_amd.agent.state import State
    script = [
        json.dumps({"mode": "multi_agent",
                    "roles": ["runbook_lookup", "ticket_history"], "waves": 1}),
        json.dumps({"tool": "final_answer", "args": {"answer": "ok finding"}}),
        json.dumps({"tool": "final_answer", "args": {"answer": "other"}}),
        json.dumps({"decision": "end", "rca": "Root cause: X."}),
    ]
    llm, ctx = _mk(store, iid, script)
    orch = Orchestrator(llm, ctx, sub_agent_turns=2)

    orig = orch._sub_agent

    async def flaky(state, role, wave, finding_id):
        if role == "ticket_history":
            raise RuntimeError("sub-agent crashed")
        return await orig(state, role, wave, finding_id)

    orch._sub_agent = flaky
    state = State(question="alert", incident_id=iid)
    res = asyncio.run(orch.run(state))
    assert "Root cause" in res.final
    by_role = {f["role"]: f for f in res.findings}
    assert "failed" in by_role["ticket_history"]["content"]


def test_orchestrator_direct_path(env):
    store, iid = env
    from aurora_amd.agent.orchestrator import Orchestrator
    from aurora_amd.agent.state import State
    script = [
        json.dumps({"mode": "direct", "roles": []}),
        json.dumps({"tool": "final_answer", "args": {"answer": "simple fix"}}),
    ]
    llm, ctx = _mk(store, iid, script)
    orch = Orchestrator(llm, ctx)
    res = asyncio.run(orch.run(State(question="minor alert", incident_id=iid)))
    assert res.final == "simple fix" and res.waves == 0


def test_mcp_server(env):
    store, iid = env
    from aurora_amd.service.mcp_server import MCPServer, search_dispatch_entries

    class RT:  # minimal runtime facade
        def __init__(self, store):
            self._store = store
            self.admin = type("A", (), {"get_user": staticmethod(
                lambda t: {"id": "u", "org_id": store.org_id} if t == "tok" else None)})()

        def org(self, org_id):
            return self._store

        def rag(self, org_id):
            from aurora_amd.rag.index import RagIndex
            return RagIndex()

        def graph(self, org_id):
            from aurora_amd.graph.infra_graph import InfraGraph
            return InfraGraph(self._store)

        class queue:  # noqa: N801
            @staticmethod
            def enqueue(*a, **k):
                return "task-1"

    srv = MCPServer(RT(store))

    async def main():
        r = await srv.handle({"jsonrpc": "2.0", "id": 1, "method": "initialize"}, "tok")
        assert r["result"]["serverInfo"]["name"] == "aurora-mi355x"
        r = await srv.handle({"jsonrpc": "2.0", "id": 2, "method": "tools/list"}, "tok")
        names = [t["name"] for t in r["result"]["tools"]]
        assert "aurora_list_incidents" in names
        assert not any(n in names for n in ("terminal_exec", "kubectl"))
        r = await srv.handle({"jsonrpc": "2.0", "id": 3, "method": "tools/call",
                              "params": {"name": "aurora_get_incident",
                                         "arguments": {"incident_id": iid}}}, "tok")
        body = json.loads(r["result"]["content"][0]["text"])
        assert body["title"] == "checkout 5xx"
        # auth required
        r = await srv.handle({"jsonrpc": "2.0", "id": 4, "method": "tools/list"}, "bad")
        assert "error" in r
    asyncio.run(main())
    assert search_dispatch_entries("postmortem")


def test_tool_confirmation_gate(env):
    """Destructive-classified tools ask the confirmer in interactive mode;
    decline cancels, headless (no confirmer) proceeds to the guardrail."""
    store, iid = env
    from aurora_amd.agent.tools.registry import GLOBAL_REGISTRY, ToolContext
    import aurora_amd.agent.tools.core  # noqa: F401

    async def main():
        asked = []

        async def deny(tool, args):
            asked.append(tool)
            return False

        ctx = ToolContext(org_store=store, confirmer=deny)
        out = await GLOBAL_REGISTRY.run(ctx, "terminal_exec", {"command": "uptime"})
        assert out.startswith("CANCELLED")
        assert asked == ["terminal_exec"]
        assert any(a["kind"] == "tool_declined" for a in store.audit_events())

        async def approve(tool, args):
            return True

        ctx2 = ToolContext(org_store=store, confirmer=approve)
        out = await GLOBAL_REGISTRY.run(ctx2, "terminal_exec", {"command": "uptime"})
        assert "load average" in out or "up" in out
        # headless: no confirmer -> guardrail path only
        ctx3 = ToolContext(org_store=store)
        out = await GLOBAL_REGISTRY.run(ctx3, "terminal_exec", {"command": "uptime"})
        assert "CANCELLED" not in out
    asyncio.run(main())


def test_skill_library_inventory():
    """Integration skills are connection-gated; core/rca always eligible
    (reference: skills/registry.py:36,430-441 — core/ + integrations/ +
    rca/ layout)."""
    from aurora_amd.agent.skills.registry import SkillRegistry
    reg = SkillRegistry()
    assert len(reg.skills) >= 20
    cats = {s.category for s in reg.skills.values()}
    assert {"core", "integrations", "rca"} <= cats
    gated = [s for s in reg.skills.values() if s.requires_connection]
    assert len(gated) >= 8
    # no connections -> no gated skill bodies in the assembled prompt
    txt = reg.assemble_rca_prompt(100000, lambda t: len(t) // 4, connected=[])
    for s in gated:
        assert f"## Skill: {s.name}" not in txt
    # rca playbooks are present and ordered by priority before generic core
    txt2 = reg.assemble_rca_prompt(100000, lambda t: len(t) // 4,
                                   connected=["datadog"])
    assert "Skill: datadog" in txt2 and "Skill: rca_deploy_rollback" in txt2


def test_roles_loaded_from_markdown():
    from aurora_amd.agent.orchestrator import ROLES, select_skills_for_role
    assert len(ROLES) >= 6
    for name, r in ROLES.items():
        assert r["brief"] and "Method" in r["brief"], name
        assert r["tools"], name
    # skill selection stays within budget and returns role-relevant text
    txt = select_skills_for_role("runbook_lookup", budget_tokens=2000)
    assert isinstance(txt, str)


def test_workflow_loop_guard(env):
    """The same tool+args repeated is intercepted (nudge at 3, forced
    final at 5) instead of burning turns (reference: sub_agent.py:81)."""
    import asyncio
    from aurora_amd.agent.state import State
    from aurora_amd.agent.workflow import Workflow
    store, iid = env
    # script: the model calls the same tool with identical args forever
    script = [json.dumps({"tool": "list_incidents", "args": {}})
              for _ in range(12)]
    llm, ctx = _mk(store, iid, script)
    wf = Workflow(llm, ctx, max_turns=10,
                  tool_subset=["list_incidents"])
    events = []

    async def main():
        async for ev in wf.stream(State(question="loop", incident_id=iid)):
            events.append(ev)
    asyncio.run(main())
    tool_calls = [e for e in events if e["type"] == "tool_call"]
    assert len(tool_calls) <= 2  # guard stopped re-execution at 3 repeats
    finals = [e for e in events if e["type"] == "final"]
    assert finals and finals[0]["content"]
    guard_msgs = [m for m in wf.__dict__.get("_last", [])] or [
        m for e in events for m in [e] if "loop guard" in str(e)]
    # the guard observation reached the context
