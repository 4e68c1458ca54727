"""Sampling parameters + constrained-decoding hooks for a generation request."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, List, Optional


@dataclass
class SamplingParams:
    temperature: float = 0.0  # 0 => greedy
    top_p: float = 1.0
    top_k: int = 0  # 0 => off
    max_tokens: int = 256
    stop: List[str] = field(default_factory=list)  # decoded-text stop strings
    stop_token_ids: List[int] = field(default_factory=list)
    seed: Optional[int] = None
    # Constrained decoding (tool-call JSON): a grammar object with
    # fork()/accepts(bytes)->bool semantics; see engine.grammar.
    grammar: Optional[Any] = None
    ignore_eos: bool = False  # synthetic benches decode a fixed budget
