"""Post-RCA artefacts: citations, fix suggestions, postmortem.

Rebuilds the reference's post-investigation stage (reference:
chat/background/citation_extractor.py:134,469; recommender.py:243-1064;
suggestion_extractor.py:53-77; postmortem_action.py)."""
from __future__ import annotations

import json
import re
from typing import List, Optional

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


RECO_PROMPT = """Based on this incident investigation, propose up to 3 fix
suggestions as a JSON object:
{{"suggestions": [{{"title": "...", "command": "<shell command or empty>",
"rationale": "..."}}]}}
Only suggest actions supported by the evidence.
Investigation:
---
{transcript}
---
JSON:"""


async def generate_recommendations(llm, store, incident_id: str,
                                   messages: List[dict]) -> List[dict]:
    """LLM suggestions -> parse -> command-safety filter -> persist
    (reference: recommender.py:839-1064; suggestion_extractor.py:53-77)."""
    try:
        res = await llm.invoke(
            RECO_PROMPT.format(transcript=render_transcript(messages, 8000)),
            SamplingParams(temperature=0.2, max_tokens=400,
                           grammar=JsonGrammar(soft_close_bytes=320)),
            purpose="rca")
        doc = json.loads(res.text)
        raw = doc.get("suggestions", [])
    except Exception:
        raw = []
    kept = []
    for s in raw[:3]:
        if not isinstance(s, dict) or not s.get("title"):
            continue
        cmd = str(s.get("command", "") or "")
        safe = not cmd or check_signature(cmd) is None
        if cmd and not safe:
            continue  # drop suggestions whose command trips the signature engine
        store.add_suggestion(incident_id, str(s["title"])[:200], cmd[:500],
                             str(s.get("rationale", ""))[:500], safe)
        kept.append(s)
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
