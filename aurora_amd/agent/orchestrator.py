"""Multi-agent orchestrator: triage -> dispatch -> parallel sub-agents ->
synthesis (reference: chat/backend/agent/orchestrator/ — triage.py:60-314,
dispatcher.py:39,220-288, sub_agent.py:81,241,268, synthesis.py:47-556,
findings_writer.py, roles/*.md).

The reference fans out with LangGraph Send and merges findings through a
state reducer; here the fan-out is asyncio tasks whose sub-agent decode
streams batch INTO THE SAME ENGINE (continuous batching — the MI355X-native
replacement for process-level parallelism), and findings
merge through the findings table.
"""
from __future__ import annotations

import asyncio
import json
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from ..engine.grammar import JsonGrammar
from ..engine.sampling_params import SamplingParams
from .state import State
from .workflow import Workflow

# role briefs (reference: orchestrator/roles/*.md)
def _load_roles() -> Dict[str, Dict[str, Any]]:
    """Role briefs live as markdown with YAML frontmatter (reference:
    orchestrator/roles/*.md + role_registry.py:45) — name, tool subset and
    skill keywords in the frontmatter, the investigation method as the
    brief body."""
    import os

    import yaml
    roles: Dict[str, Dict[str, Any]] = {}
    d = os.path.join(os.path.dirname(__file__), "roles")
    if not os.path.isdir(d):
        return roles
    for fn in sorted(os.listdir(d)):
        if not fn.endswith(".md"):
            continue
        with open(os.path.join(d, fn)) as f:
            raw = f.read()
        body = raw
        meta: Dict[str, Any] = {}
        if raw.startswith("---"):
            _, fm, body = raw.split("---", 2)
            meta = yaml.safe_load(fm) or {}
        name = meta.get("name") or fn[:-3]
        roles[name] = {
            "brief": body.strip(),
            "tools": list(meta.get("tools") or []),
            "skills": list(meta.get("skills") or []),
        }
    return roles


ROLES: Dict[str, Dict[str, Any]] = _load_roles()


def select_skills_for_role(role: str, budget_tokens: int = 1200,
                           count_tokens=None) -> str:
    """Per-role skill selection (reference: orchestrator/select_skills.py):
    skills whose name/tags overlap the role's keywords, under a budget."""
    from .skills.registry import get_skill_registry
    keywords = set(ROLES.get(role, {}).get("skills", []))
    keywords |= set(role.split("_"))
    reg = get_skill_registry()
    count = count_tokens or (lambda t: len(t) // 4)
    picked = []
    used = 0
    for sk in sorted(reg.skills.values(), key=lambda s: s.rca_priority):
        hay = (sk.name + " " + sk.description + " " + sk.category).lower()
        if not any(k in hay for k in keywords):
            continue
        cost = count(sk.body)
        if used + cost > budget_tokens:
            continue
        picked.append(f"## skill: {sk.name}\n{sk.body}")
        used += cost
    return "\n\n".join(picked)

TRIAGE_PROMPT = """You are the RCA orchestrator. Given the alert, pick the
investigation plan as a JSON object:
{{"mode": "direct" | "multi_agent", "roles": ["<role>", ...], "waves": 1}}
Valid roles: {roles}. Use "direct" for simple/low-severity alerts; pick 2-4
roles for complex ones.
Alert: {alert}
JSON:"""

SYNTH_PROMPT = """You are the synthesis agent. Merge the sub-agent findings
below into either a follow-up plan or a final RCA. Reply a JSON object:
{{"decision": "end" | "dispatch", "roles": [...], "rca": "<final RCA text if
decision is end>"}}
Findings:
{findings}
JSON:"""


@dataclass
class OrchestratorResult:
    final: str
    waves: int
    findings: List[dict] = field(default_factory=list)
    triage: Optional[dict] = None


class Orchestrator:
    def __init__(self, llm, ctx, max_waves: int = 2, sub_agent_turns: int = 6,
                 sub_agent_timeout_s: float = 600.0, max_roles_per_wave: int = 4):
        self.llm = llm
        self.ctx = ctx
        self.max_waves = max_waves
        self.sub_agent_turns = sub_agent_turns
        self.sub_agent_timeout_s = sub_agent_timeout_s
        self.max_roles = max_roles_per_wave

    async def run(self, state: State) -> OrchestratorResult:
        store = self.ctx.org_store
        alert = state.question[:3000]
        # ---- triage (reference: triage.py:54-71) ----
        triage = await self._json_call(
            TRIAGE_PROMPT.format(roles=sorted(ROLES), alert=alert))
        roles = self._filter_roles(triage.get("roles", []))
        state.triage_decision = triage
        if triage.get("mode") != "multi_agent" or not roles:
            # direct ReAct path (reference: workflow.py:171-205 direct_react)
            wf = Workflow(self.llm, self.ctx, max_turns=self.sub_agent_turns * 2)
            final = ""
            async for ev in wf.stream(state):
                if ev["type"] == "final":
                    final = ev["content"]
            return OrchestratorResult(final, 0, triage=triage)

        all_findings: List[dict] = []
        final = ""
        for wave in range(self.max_waves):
            state.synthesis_wave = wave
            # ---- dispatch: pre-emit finding rows then fan out
            # (reference: dispatcher.py:220-288) ----
            fids = {}
            for role in roles:
                fids[role] = store.upsert_finding(
                    state.incident_id or "-", role, wave, "running") \
                    if store is not None else None
            tasks = [self._sub_agent(state, role, wave, fids[role])
                     for role in roles]
            results = await asyncio.gather(*tasks, return_exceptions=True)
            wave_findings = []
            for role, res in zip(roles, results):
                if isinstance(res, Exception):
                    res = f"(sub-agent {role} failed: {res})"
                    if store is not None and fids[role]:
                        store.upsert_finding(state.incident_id or "-", role,
                                             wave, "failed", str(res),
                                             finding_id=fids[role])
                wave_findings.append({"role": role, "wave": wave,
                                      "content": str(res)[:4000]})
            all_findings.extend(wave_findings)
            state.finding_refs.extend(f["role"] for f in wave_findings)
            # ---- synthesis (reference: synthesis.py:47-61,556) ----
            synth = await self._json_call(SYNTH_PROMPT.format(
                findings=json.dumps(all_findings)[:8000]))
            state.synthesis_history.append(json.dumps(synth)[:500])
            if synth.get("decision") == "dispatch" and wave + 1 < self.max_waves:
                roles = self._filter_roles(synth.get("roles", [])) or roles
                continue
            final = str(synth.get("rca", "")).strip()
            break
        if not final:
            final = "Root cause undetermined. Sub-agent findings:\n" + "\n".join(
                f"- [{f['role']}] {f['content'][:300]}" for f in all_findings)
        return OrchestratorResult(final, state.synthesis_wave + 1,
                                  findings=all_findings, triage=triage)

    # ------------------------------------------------------------- helpers

    def _filter_roles(self, roles: List[str]) -> List[str]:
        """known-role filter + dedupe + cap (reference: dispatcher.py:69-122)."""
        out: List[str] = []
        for r in roles:
            if r in ROLES and r not in out:
                out.append(r)
        return out[: self.max_roles]

    async def _json_call(self, prompt: str) -> dict:
        try:
            res = await self.llm.invoke(
                prompt, SamplingParams(temperature=0.1, max_tokens=220,
                                       grammar=JsonGrammar(soft_close_bytes=220)),
                purpose="orchestrator")
            doc = json.loads(res.text)
            return doc if isinstance(doc, dict) else {}
        except Exception:  # noqa: BLE001
            return {}

    async def _sub_agent(self, state: State, role: str, wave: int,
                         finding_id: Optional[str]) -> str:
        """Role-briefed ReAct run with a tool subset + timeout (reference:
        sub_agent.py:81,241,268); the finding is persisted to the findings
        table (findings_writer.py)."""
        brief = ROLES[role]["brief"]
        subset = ROLES[role]["tools"] + ["final_answer"]
        # brief lives in the system override only — duplicating it in the
        # question would re-prefill it per turn as history
        sub_state = State(
            question=f"[role: {role}] Incident: {state.question}",
            user_id=state.user_id, org_id=state.org_id,
            session_id=None,  # sub-agents keep their own scratch context
            incident_id=state.incident_id, is_background=True)
        skills_text = select_skills_for_role(
            role, count_tokens=getattr(self.llm, "count_tokens", None))
        wf = Workflow(self.llm, self.ctx, tool_subset=subset,
                      system_prompt_override=(
                          f"You are the {role} sub-agent of an incident "
                          f"investigation.\n{brief}\n"
                          + (f"\n# Relevant skills\n{skills_text}\n"
                             if skills_text else "")
                          + "Finish with final_answer containing your "
                            "findings."),
                      max_turns=self.sub_agent_turns)
        final = ""
        try:
            async def _run():
                nonlocal final
                async for ev in wf.stream(sub_state):
                    if ev["type"] == "final":
                        final = ev["content"]
            await asyncio.wait_for(_run(), timeout=self.sub_agent_timeout_s)
        except asyncio.TimeoutError:
            final = final or "(sub-agent timed out)"
        store = self.ctx.org_store
        if store is not None and finding_id:
            store.upsert_finding(state.incident_id or "-", role, wave,
                                 "completed", final, finding_id=finding_id)
        return final
