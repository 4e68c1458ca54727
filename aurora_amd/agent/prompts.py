"""Segmented system-prompt composer (reference: chat/backend/agent/prompt/
composer.py — build_prompt_segments/assemble_system_prompt; stable segments
first so the engine's session-KV prefix reuse gets maximal hits, the
in-process analog of the reference's vendor prompt-prefix caching,
utils/prefix_cache.py:39-158)."""
from __future__ import annotations

import json
from typing import List

from .state import State
from .tools.registry import Tool


def action_protocol() -> str:
    return (
        "Respond in repeating Thought/Action steps.\n"
        "After each 'Thought:' line, an 'Action:' line follows containing ONE "
        'JSON object: {"tool": "<tool name>", "args": {...}}.\n'
        'To finish, use {"tool": "final_answer", "args": {"answer": "<your RCA>"}}.\n'
        "Observations from tools are appended after each action."
    )


def build_prompt_segments(state: State, tools: List[Tool], skills_text: str,
                          alert_context: str = "") -> List[str]:
    segs = [skills_text or "You are Aurora, an incident-investigation agent."]
    segs.append("# Tools\n" + "\n".join(t.spec_line() for t in tools)
                + "\n- final_answer(answer: the final RCA report): finish the "
                  "investigation")
    segs.append("# Protocol\n" + action_protocol())
    if state.mode == "ask":
        segs.append("# Mode\nRead-only (ask) mode: you may only inspect, never change.")
    if alert_context:
        segs.append("# Alert context\n" + alert_context)
    if state.rca_context:
        segs.append("# Prior investigation context\n" + state.rca_context)
    if state.incident_start_time:
        segs.append(f"# Incident start time (unix)\n{state.incident_start_time}")
    return segs


def assemble_system_prompt(segs: List[str]) -> str:
    return "\n\n".join(segs)


def render_history(messages: List[dict], max_messages: int = 24,
                   tool_output_chars: int = 1600) -> str:
    """Flatten the conversation for the next completion; last-N message
    window + per-observation truncation (reference: agent.py:86,662-664,
    691-692 history cleanup). The reference's 4 KB ToolMessage cap is a
    ~1k-BPE-token budget; with the byte tokenizer (1 token/byte) the
    equivalent information budget is ~1.6 KB — 4000 here quadrupled every
    prompt, prefill and KV footprint for the same content."""
    parts: List[str] = []
    for m in messages[-max_messages:]:
        role = m.get("role")
        if role == "user":
            parts.append(f"User: {m['content']}")
        elif role == "thought":
            parts.append(f"Thought: {m['content']}")
        elif role == "action":
            parts.append("Action: " + json.dumps(
                {"tool": m.get("tool"), "args": m.get("args", {})}))
        elif role == "observation":
            out = m.get("content", "")
            if len(out) > tool_output_chars:
                out = out[:tool_output_chars] + "...[truncated]"
            parts.append(f"Observation: {out}")
        elif role == "assistant":
            parts.append(f"Assistant: {m['content']}")
    return "\n".join(parts)
