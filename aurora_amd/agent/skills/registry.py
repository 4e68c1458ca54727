"""Skill registry: markdown playbooks with YAML frontmatter, ordered by
rca_priority under a token budget (reference: chat/backend/agent/skills/
registry.py:36,430-441; skills are core/ + integrations/<name>/SKILL.md +
rca/ in the reference — same layout here)."""
from __future__ import annotations

import os
import re
from dataclasses import dataclass
from typing import Dict, List, Optional

import yaml

SKILLS_DIR = os.path.join(os.path.dirname(__file__), "library")


@dataclass
class Skill:
    name: str
    description: str
    body: str
    category: str = "core"
    rca_priority: int = 100
    requires_connection: Optional[str] = None


class SkillRegistry:
    def __init__(self, path: str = SKILLS_DIR):
        self.skills: Dict[str, Skill] = {}
        if os.path.isdir(path):
            for root, _dirs, files in os.walk(path):
                for fn in files:
                    if fn.endswith(".md"):
                        self._load(os.path.join(root, fn),
                                   os.path.relpath(root, path))

    def _load(self, fp: str, category: str) -> None:
        text = open(fp).read()
        meta: dict = {}
        body = text
        m = re.match(r"^---\n(.*?)\n---\n(.*)$", text, re.DOTALL)
        if m:
            meta = yaml.safe_load(m.group(1)) or {}
            body = m.group(2)
        name = meta.get("name") or os.path.splitext(os.path.basename(fp))[0]
        self.skills[name] = Skill(
            name=name,
            description=meta.get("description", ""),
            body=body.strip(),
            category=category if category != "." else "core",
            rca_priority=int(meta.get("rca_priority", 100)),
            requires_connection=meta.get("requires_connection"),
        )

    def get(self, name: str) -> Optional[Skill]:
        return self.skills.get(name)

    def catalog(self) -> List[dict]:
        return [{"name": s.name, "description": s.description,
                 "category": s.category} for s in self.skills.values()]

    def assemble_rca_prompt(self, token_budget: int, count_tokens,
                            connected: Optional[List[str]] = None) -> str:
        """Concatenate skills by rca_priority until the budget is spent
        (reference: registry.py:430-441), skipping connection-gated skills
        whose connector is absent."""
        parts: List[str] = []
        used = 0
        for s in sorted(self.skills.values(), key=lambda s: s.rca_priority):
            if s.requires_connection and (connected is None or
                                          s.requires_connection not in connected):
                continue
            cost = count_tokens(s.body)
            if used + cost > token_budget:
                continue
            used += cost
            parts.append(f"## Skill: {s.name}\n{s.body}")
        return "\n\n".join(parts)


_REG: Optional[SkillRegistry] = None


def get_skill_registry() -> SkillRegistry:
    global _REG
    if _REG is None:
        _REG = SkillRegistry()
    return _REG
