"""Sigma-style command signature engine.

The reference ships 37 SigmaHQ Linux process-creation rules compiled to
regex at import (reference: utils/security/sigma_loader.py:241-292,
signature_match.py:128).  Same architecture here: YAML rules in
``rules/*.yml`` with a small detection grammar (contains / startswith /
regex over the command line, AND of selections, `condition: selection`),
compiled once to Python regex; matching is pure host-side work.
"""
from __future__ import annotations

import os
import re
from dataclasses import dataclass
from typing import List, Optional

import yaml

RULES_DIR = os.path.join(os.path.dirname(__file__), "rules")


@dataclass
class SigmaRule:
    rule_id: str
    title: str
    level: str
    patterns: List[re.Pattern]  # ALL must match (AND within the rule)

    def matches(self, command: str) -> bool:
        return all(p.search(command) for p in self.patterns)


@dataclass
class SignatureMatch:
    rule_id: str
    title: str
    level: str


def _compile_value(v: str, mode: str) -> re.Pattern:
    if mode == "re":
        return re.compile(v, re.IGNORECASE)
    esc = re.escape(v)
    if mode == "startswith":
        return re.compile(r"^\s*" + esc, re.IGNORECASE)
    return re.compile(esc, re.IGNORECASE)  # contains


def load_rules(path: str = RULES_DIR) -> List[SigmaRule]:
    rules: List[SigmaRule] = []
    if not os.path.isdir(path):
        return rules
    for fn in sorted(os.listdir(path)):
        if not fn.endswith((".yml", ".yaml")):
            continue
        with open(os.path.join(path, fn)) as f:
            doc = yaml.safe_load(f)
        if not doc:
            continue
        det = doc.get("detection", {})
        pats: List[re.Pattern] = []
        for key, spec in det.items():
            if key == "condition":
                continue
            if isinstance(spec, dict):
                for field, vals in spec.items():
                    mode = "contains"
                    if field.endswith("|re"):
                        mode = "re"
                    elif field.endswith("|startswith"):
                        mode = "startswith"
                    if isinstance(vals, str):
                        vals = [vals]
                    # list of values = OR -> one alternation regex
                    subs = [_compile_value(v, mode).pattern for v in vals]
                    pats.append(re.compile("|".join(f"(?:{s})" for s in subs),
                                           re.IGNORECASE))
        if pats:
            rules.append(SigmaRule(
                rule_id=doc.get("id", fn),
                title=doc.get("title", fn),
                level=doc.get("level", "high"),
                patterns=pats,
            ))
    return rules


_RULES: Optional[List[SigmaRule]] = None


def rules() -> List[SigmaRule]:
    global _RULES
    if _RULES is None:
        _RULES = load_rules()
    return _RULES


def check_signature(command: str) -> Optional[SignatureMatch]:
    """First matching rule, or None (reference: signature_match.py:128)."""
    for r in rules():
        if r.matches(command):
            return SignatureMatch(r.rule_id, r.title, r.level)
    return None
