"""Investigation state carried through the workflow (reference:
chat/backend/agent/utils/state.py:8-61)."""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


@dataclass
class State:
    question: str
    user_id: Optional[str] = None
    org_id: Optional[str] = None
    session_id: Optional[str] = None
    incident_id: Optional[str] = None
    incident_start_time: Optional[float] = None
    mode: str = "agent"  # "agent" | "ask" (read-only)
    is_background: bool = False
    rca_context: Optional[str] = None
    provider_preference: Optional[str] = None
    guardrail_blocked: bool = False
    permitted_tools: Optional[List[str]] = None
    # conversation so far: list of {"role": ..., "content": ...,
    # optional "tool"/"args"/"output"}
    messages: List[Dict[str, Any]] = field(default_factory=list)
    # orchestrator fields (reference: state.py:53 finding_refs reducer)
    triage_decision: Optional[Dict[str, Any]] = None
    subagent_inputs: List[Dict[str, Any]] = field(default_factory=list)
    finding_refs: List[str] = field(default_factory=list)
    synthesis_wave: int = 0
    synthesis_history: List[str] = field(default_factory=list)
    created_at: float = field(default_factory=time.time)
