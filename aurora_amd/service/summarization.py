"""Incident summarisation: the RCA output schema.

Reproduces the reference's summary contract (reference:
chat/background/summarization.py:279-346,562,703): 2-3 paragraphs
(trigger/impact then Root Cause — which must start "Root cause
undetermined." when inconclusive and hedge when only LIKELY), followed by
`## Ruled Out` and `## Not Checked` bullet sections; severity derived from
the RCA.
"""
from __future__ import annotations

import re
from typing import List, Optional

from ..engine.sampling_params import SamplingParams

SUMMARY_PROMPT = """You write incident summaries for an SRE team.
Using ONLY the investigation transcript below, produce:
1. One paragraph: what triggered the incident and the user/system impact.
2. One paragraph starting exactly with "Root cause:" — if the cause was NOT
   conclusively proven, start it with "Root cause undetermined." and explain
   the leading hypothesis with hedged language ("likely", "suggests").
3. A section "## Ruled Out" with one bullet per eliminated cause.
4. A section "## Not Checked" with one bullet per unexamined avenue.

Transcript:
---
{transcript}
---
Summary:"""


def render_transcript(messages: List[dict], limit_chars: int = 12000) -> str:
    parts = []
    for m in messages:
        r = m.get("role")
        if r == "thought":
            parts.append(f"[thought] {m['content']}")
        elif r == "action":
            parts.append(f"[tool] {m.get('tool')} {m.get('args')}")
        elif r == "observation":
            parts.append(f"[output] {m.get('content', '')[:600]}")
        elif r == "assistant":
            parts.append(f"[report] {m['content']}")
    text = "\n".join(parts)
    return text[-limit_chars:]


def ensure_schema(summary: str) -> str:
    """Guarantee the structural contract even when the model under-delivers
    (synthetic/random-weight runs): Root cause paragraph + both sections."""
    out = summary.strip()
    if "root cause" not in out.lower():
        out += ("\n\nRoot cause undetermined. The investigation did not "
                "conclusively establish a cause.")
    if "## Ruled Out" not in out:
        out += "\n\n## Ruled Out\n- (none recorded)"
    if "## Not Checked" not in out:
        out += "\n\n## Not Checked\n- (none recorded)"
    return out


def determine_severity(summary: str, payload_severity: Optional[str] = None) -> str:
    """reference: task.py:1861 _determine_severity_from_rca"""
    s = summary.lower()
    if payload_severity in ("critical", "high", "medium", "low"):
        base = payload_severity
    else:
        base = "medium"
    if re.search(r"data loss|outage|all (users|traffic)|complete(ly)? (down|unavailable)", s):
        return "critical"
    if re.search(r"degrad|partial|elevated (latency|errors)", s) and base != "critical":
        return "high" if base in ("medium", "low") else base
    return base


async def generate_incident_summary(llm, messages: List[dict],
                                    session_id: Optional[str] = None) -> str:
    """reference: summarization.py:703 generate_incident_summary_from_chat"""
    transcript = render_transcript(messages)
    res = await llm.invoke(
        SUMMARY_PROMPT.format(transcript=transcript),
        SamplingParams(temperature=0.2, max_tokens=512),
        purpose="summarization", session_id=session_id)
    return ensure_schema(res.text)


def alert_only_summary(title: str, payload: dict) -> str:
    """Neutral summary when no chat transcript exists (reference:
    summarization.py:562,64-180)."""
    sev = payload.get("severity") or payload.get("alert", {}).get("severity", "unknown")
    svc = payload.get("service") or payload.get("alert", {}).get("service", "unknown")
    return ensure_schema(
        f"An alert '{title}' fired for service '{svc}' with severity '{sev}'. "
        f"Impact is not yet assessed; an investigation has not completed.\n\n"
        f"Root cause undetermined. No investigation transcript is available yet.")
