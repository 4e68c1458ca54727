"""Post-RCA artefacts: citations, fix suggestions, postmortem.

Rebuilds the reference's post-investigation stage (reference:
chat/background/citation_extractor.py:134,469; recommender.py:243-1064;
suggestion_extractor.py:53-77; postmortem_action.py)."""
from __future__ import annotations

import json
import re
from typing import List

from ..engine.grammar import JsonGrammar
from ..engine.sampling_params import SamplingParams
from ..guardrails.sigma import check_signature
from .summarization import render_transcript


def extract_citations(messages: List[dict]) -> List[dict]:
    """Per-claim citations from the tool trace: each observation that the
    final report plausibly drew on becomes a citation row (reference:
    citation_extractor.py:134 — trace-driven, not model-claimed)."""
    final = next((m["content"] for m in reversed(messages)
                  if m.get("role") == "assistant"), "")
    cits: List[dict] = []
    last_tool = None
    for m in messages:
        if m.get("role") == "action":
            last_tool = m.get("tool")
        elif m.get("role") == "observation" and last_tool:
            out = (m.get("content") or "").strip()
            if not out or out.startswith("ERROR"):
                continue
            # claim = the sentence of the final report sharing the most
            # informative words with this observation
            best, best_score = None, 0
            obs_words = set(re.findall(r"[a-z0-9_\-]{4,}", out.lower()))
            for sent in re.split(r"(?<=[.!?])\s+", final):
                w = set(re.findall(r"[a-z0-9_\-]{4,}", sent.lower()))
                score = len(w & obs_words)
                if score > best_score:
                    best, best_score = sent, score
            if best and best_score >= 2:
                cits.append({"claim": best.strip(), "source_tool": last_tool,
                             "evidence": out[:800]})
    return cits


def save_citations(store, incident_id: str, messages: List[dict]) -> int:
    cits = extract_citations(messages)
    for c in cits:
        store.add_citation(incident_id, c["claim"], c["source_tool"], c["evidence"])
    return len(cits)


# --------------------------------------------------- hypothesis ledger
# (reference: recommender.py:243-353 — structured InvestigationState
# extracted from the agent's reasoning, fed to the recommendation prompt)

HYPOTHESIS_PROMPT = """Extract the hypothesis state from this investigation
reasoning. Reply ONLY a JSON object:
{{"hypotheses": [{{"statement": "...", "probability": "high|medium|low",
"status": "confirmed|open",
"mitigation_class": "rollback|restart|scale|config_revert|code_fix|unknown"}}],
"ruled_out": ["..."], "unexplored": ["..."]}}
REASONING:
---
{reasoning}
---
JSON:"""

_HYPOTHESIS_REASONING_CHARS = 6000


def _reasoning_slice(text: str) -> str:
    """Keep BOTH ends when over budget: hypothesis formation lands early,
    the confirmed root cause late (reference: recommender.py:255-264)."""
    if len(text) <= _HYPOTHESIS_REASONING_CHARS:
        return text
    head = _HYPOTHESIS_REASONING_CHARS // 2
    tail = _HYPOTHESIS_REASONING_CHARS - head
    return (text[:head] + "\n...[middle of reasoning truncated]...\n"
            + text[-tail:])


async def extract_hypotheses(llm, messages: List[dict]) -> dict:
    """Hypothesis ledger from the agent's thoughts. Empty ledger on any
    failure (the recommender prompt degrades gracefully)."""
    reasoning = "\n".join(m.get("content", "") for m in messages
                          if m.get("role") == "thought")
    empty = {"hypotheses": [], "ruled_out": [], "unexplored": []}
    if len(reasoning.strip()) < 80:
        return empty
    try:
        res = await llm.invoke(
            HYPOTHESIS_PROMPT.format(reasoning=_reasoning_slice(reasoning)),
            SamplingParams(temperature=0.1, max_tokens=400,
                           grammar=JsonGrammar(soft_close_bytes=320)),
            purpose="rca")
        doc = json.loads(res.text)
    except Exception:  # noqa: BLE001
        return empty
    out = dict(empty)
    hs = []
    for h in (doc.get("hypotheses") or [])[:8]:
        if isinstance(h, dict) and h.get("statement"):
            hs.append({
                "statement": str(h["statement"])[:300],
                "probability": str(h.get("probability", "medium")),
                "status": str(h.get("status", "open")),
                "mitigation_class": str(h.get("mitigation_class", "unknown")),
            })
    out["hypotheses"] = hs
    out["ruled_out"] = [str(r)[:300] for r in (doc.get("ruled_out") or [])[:8]
                        if isinstance(r, str)]
    out["unexplored"] = [str(u)[:300] for u in (doc.get("unexplored") or [])[:8]
                         if isinstance(u, str)]
    return out


def format_ledger(ledger: dict) -> str:
    if not (ledger.get("hypotheses") or ledger.get("ruled_out")):
        return ""
    lines = ["HYPOTHESIS LEDGER:"]
    for h in ledger.get("hypotheses", []):
        tag = "CONFIRMED" if h.get("status") == "confirmed" else h.get(
            "probability", "medium")
        lines.append(f"- [{tag}] {h['statement']} "
                     f"(mitigation: {h.get('mitigation_class', 'unknown')})")
    for r in ledger.get("ruled_out", []):
        lines.append(f"- [ruled out] {r}")
    for u in ledger.get("unexplored", []):
        lines.append(f"- [unexplored] {u}")
    return "\n".join(lines)


# ------------------------------------------- safe self-run diagnostics
# (reference: recommender.py:503-609 — plan additional read-only commands
# the investigation missed, run them, fold the evidence in)

DIAG_PROMPT = """The investigation already ran these commands:
{executed}
What additional SAFE, READ-ONLY diagnostic commands would add useful
evidence? Reply ONLY a JSON object:
{{"commands": [{{"command": "...", "rationale": "..."}}]}}
Max 3; [] if the investigation is thorough.
JSON:"""

MAX_SELF_EXEC = 3


async def self_execute_diagnostics(llm, ctx, messages: List[dict]) -> List[dict]:
    """Plan + run extra read-only diagnostics through the SAME guarded
    execution path the agent uses (signature -> policy -> judge; anything
    not provably read-only is dropped, never run)."""
    executed = []
    for m in messages:
        if m.get("role") == "action" and m.get("tool") in (
                "terminal_exec", "kubectl", "cloud_exec"):
            args = m.get("args") or {}
            executed.append(str(args.get("command") or args.get("args") or ""))
    try:
        res = await llm.invoke(
            DIAG_PROMPT.format(executed="\n".join(executed[-10:]) or "(none)"),
            SamplingParams(temperature=0.1, max_tokens=300,
                           grammar=JsonGrammar(soft_close_bytes=230)),
            purpose="rca")
        plan = json.loads(res.text).get("commands", [])
    except Exception:  # noqa: BLE001
        return []
    from ..guardrails.command_safety import _heuristic_readonly
    ran: List[dict] = []
    seen = {e.strip().lower() for e in executed}
    for c in plan[:MAX_SELF_EXEC]:
        cmd = str((c or {}).get("command", "")).strip()
        if not cmd or cmd.lower() in seen:
            continue
        # stricter than the agent path: only the read-only allowlist runs
        # unattended post-RCA (no judge override)
        if check_signature(cmd) is not None or not _heuristic_readonly(cmd):
            continue
        from ..agent.tools.registry import GLOBAL_REGISTRY
        out = await GLOBAL_REGISTRY.run(ctx, "terminal_exec",
                                        {"command": cmd}, timeout_s=20.0)
        ran.append({"command": cmd, "rationale": str(c.get("rationale", ""))[:200],
                    "output": out[:600]})
    return ran


def validate_inventory(suggestions: List[dict], known_resources) -> List[dict]:
    """Drop suggestions whose command names resources we have never seen
    (reference: recommender.py:1024 — no hallucinated targets)."""
    known = {str(r).lower() for r in known_resources}
    kept = []
    for s in suggestions:
        cmd = str(s.get("command", "") or "")
        # resource-looking tokens: kubectl/helm/aws object names
        m = re.findall(r"(?:deployment|service|pod|statefulset|node|svc|deploy)"
                       r"[/ ]([a-z0-9.-]{2,40})", cmd.lower())
        if m and not any(tok in known for tok in m):
            s = dict(s, command="",
                     rationale=str(s.get("rationale", ""))
                     + " [command dropped: references a resource not in "
                       "the inventory]")
        kept.append(s)
    return kept


RECO_PROMPT = """Based on this incident investigation, propose up to 3 fix
suggestions as a JSON object:
{{"suggestions": [{{"title": "...", "command": "<shell command or empty>",
"rationale": "..."}}]}}
Only suggest actions supported by the evidence.
{ledger}
{diagnostics}
Investigation:
---
{transcript}
---
JSON:"""


async def generate_recommendations(llm, store, incident_id: str,
                                   messages: List[dict],
                                   ctx=None) -> List[dict]:
    """Hypothesis ledger -> optional safe self-run diagnostics -> LLM
    suggestions -> parse -> safety + inventory filter -> persist
    (reference: recommender.py:243-1064; suggestion_extractor.py:53-77)."""
    # hypothesis extraction and diagnostics planning both depend only on
    # the transcript — overlap their LLM passes to keep the decode batch full
    import asyncio as _aio
    import os
    ledger_task = _aio.create_task(extract_hypotheses(llm, messages))
    diags: List[dict] = []
    if (ctx is not None
            and os.environ.get("AURORA_SELF_DIAGNOSTICS", "1") != "0"):
        diags = await self_execute_diagnostics(llm, ctx, messages)
    ledger = await ledger_task
    diag_text = ""
    if diags:
        diag_text = "ADDITIONAL DIAGNOSTICS RUN POST-INVESTIGATION:\n" + \
            "\n".join(f"$ {d['command']}\n{d['output'][:300]}" for d in diags)
    try:
        res = await llm.invoke(
            RECO_PROMPT.format(transcript=render_transcript(messages, 8000),
                               ledger=format_ledger(ledger),
                               diagnostics=diag_text),
            SamplingParams(temperature=0.2, max_tokens=400,
                           grammar=JsonGrammar(soft_close_bytes=320)),
            purpose="rca")
        doc = json.loads(res.text)
        raw = doc.get("suggestions", [])
    except Exception:
        raw = []
    # inventory: graph services + resources named in the tool trace
    known = set()
    if ctx is not None and getattr(ctx, "graph", None) is not None:
        try:
            known |= {s.get("name", "").lower()
                      for s in ctx.graph.list_services()}
        except Exception:  # noqa: BLE001
            pass
    for m in messages:
        if m.get("role") == "observation":
            known |= set(re.findall(r"[a-z0-9][a-z0-9.-]{2,40}",
                                    (m.get("content") or "")[:2000].lower()))
    raw = validate_inventory([s for s in raw if isinstance(s, dict)], known)
    kept = []
    for s in raw[:3]:
        if not s.get("title"):
            continue
        cmd = str(s.get("command", "") or "")
        safe = not cmd or check_signature(cmd) is None
        if cmd and not safe:
            continue  # drop suggestions whose command trips the signature engine
        store.add_suggestion(incident_id, str(s["title"])[:200], cmd[:500],
                             str(s.get("rationale", ""))[:500], safe)
        kept.append(s)
    if ledger.get("hypotheses") or ledger.get("ruled_out"):
        store.save_artifact(f"hypothesis_ledger:{incident_id}",
                            json.dumps(ledger))
    return kept


def build_postmortem(incident: dict, summary: str, citations: List[dict],
                     suggestions: List[dict], timeline: List[dict]) -> str:
    """Deterministic postmortem assembly from the investigation record
    (auto-generated postmortem — reference: postmortem_action.py)."""
    import datetime as _dt

    def _iso(ts) -> str:
        try:
            return _dt.datetime.utcfromtimestamp(float(ts)).strftime(
                "%Y-%m-%d %H:%M:%SZ")
        except (TypeError, ValueError, OSError, OverflowError):
            return str(ts)

    status = incident.get("aurora_status") or incident.get("status") or "open"
    lines = [
        f"# Postmortem: {incident['title']}",
        "",
        f"- **Incident**: {incident['id']}",
        f"- **Source**: {incident['source']}",
        f"- **Severity**: {incident.get('severity') or 'unknown'}",
        f"- **Opened**: {_iso(incident['created_at'])}",
        f"- **Investigation**: {status}",
        "",
        "## Summary",
        summary,
        "",
        "## Timeline",
    ]
    for ev in timeline:
        lines.append(f"- {_iso(ev['ts'])}: {ev['event']}")
    if citations:
        lines += ["", "## Evidence"]
        for c in citations[:10]:
            lines.append(f"- **{c['source_tool']}**: {c['claim']}")
    if suggestions:
        lines += ["", "## Follow-ups"]
        for s in suggestions:
            lines.append(f"- {s.get('title')}: {s.get('rationale', '')}")
    lines += ["", "## Lessons",
              "- (fill in during postmortem review)"]
    return "\n".join(lines)
