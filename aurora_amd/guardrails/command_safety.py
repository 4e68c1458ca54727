"""Guarded command execution: signature engine first, then the LLM judge.

Mirrors the reference's single source of truth `evaluate_command`
(reference: utils/security/command_safety.py:313-367 — check_signature then
check_command_safety, fail-closed) with the judge running on the in-process
engine under a JSON grammar instead of a rented model.
"""
from __future__ import annotations

import json
import time
from dataclasses import dataclass
from typing import Optional

from ..engine.grammar import JsonGrammar
from ..engine.sampling_params import SamplingParams
from .sigma import check_signature

JUDGE_PROMPT = """You are a security judge for infrastructure commands run by an
incident-investigation agent. Reply with ONLY a JSON object:
{{"safe": true|false, "reason": "<short>"}}
A command is unsafe if it modifies or destroys infrastructure, exfiltrates
secrets, or is not a read-only diagnostic.
Command: {command}
JSON:"""


@dataclass
class SafetyVerdict:
    safe: bool
    reason: str
    source: str  # "signature" | "llm" | "policy" | "error"
    rule_id: Optional[str] = None
    latency_ms: float = 0.0


# conservative allowlist of clearly read-only diagnostics: lets the
# heuristic layer pass obvious cases without an LLM round-trip
_READONLY_PREFIXES = (
    "ls", "cat /var/log", "cat /proc", "ps", "top -b", "df", "du", "free",
    "uptime", "uname", "id", "whoami", "date", "hostname", "ip addr",
    "ip route", "ss ", "netstat", "dig ", "nslookup", "ping -c",
    "kubectl get", "kubectl describe", "kubectl logs", "kubectl top",
    "kubectl api-resources", "kubectl explain", "kubectl version",
    "aws ec2 describe", "aws s3 ls", "aws logs", "aws cloudwatch",
    "aws sts get-caller-identity", "gcloud compute instances list",
    "gcloud logging read", "az vm list", "az monitor", "journalctl",
    "systemctl status", "git log", "git show", "git diff", "helm list",
    "helm status", "docker ps", "docker logs", "docker inspect",
    "grep ", "zgrep ", "egrep ", "tail ", "head ", "wc ", "stat ",
    "dmesg", "lsof", "vmstat", "iostat", "mpstat", "nproc", "file ",
    "find /var/log", "find /tmp",
)


#: Any of these inside a command means it is (or could expand into) more than
#: one command / an output redirection, so the read-only fast path must not
#: apply: the LLM judge sees it instead. Single '&', newline and carriage
#: return are included — 'kubectl get pods\nrm -rf /' is two commands.
_SHELL_META = ("|", ";", "&", ">", "<", "`", "$(", "\n", "\r")


def _heuristic_readonly(command: str) -> bool:
    c = command.strip()
    if any(ch in c for ch in _SHELL_META):
        return False
    return any(c.startswith(p) for p in _READONLY_PREFIXES)


async def evaluate_command(command: str, llm=None, org_store=None,
                           user_id: Optional[str] = None) -> SafetyVerdict:
    """signature -> org policy -> read-only heuristic -> LLM judge.
    Fails CLOSED: any judge failure blocks (reference: command_safety.py:332)."""
    t0 = time.monotonic()

    def _ms() -> float:
        return (time.monotonic() - t0) * 1000.0

    sig = check_signature(command)
    if sig is not None:
        v = SafetyVerdict(False, f"matched signature rule: {sig.title}",
                          "signature", rule_id=sig.rule_id, latency_ms=_ms())
        if org_store is not None:
            org_store.audit("command_blocked",
                            json.dumps({"command": command[:500], "rule": sig.rule_id}))
        return v

    # org command policies (reference: utils/auth/command_gate.py:306):
    # deny rules veto; allow rules short-circuit the judge
    if org_store is not None:
        import re as _re
        for pol in org_store.command_policies():
            try:
                if _re.search(pol["pattern"], command):
                    if pol["action"] == "deny":
                        org_store.audit("command_blocked", json.dumps(
                            {"command": command[:500], "policy": pol["id"]}))
                        return SafetyVerdict(False, "org policy deny", "policy",
                                             rule_id=pol["id"], latency_ms=_ms())
                    return SafetyVerdict(True, "org policy allow", "policy",
                                         rule_id=pol["id"], latency_ms=_ms())
            except _re.error:
                continue

    if _heuristic_readonly(command):
        return SafetyVerdict(True, "read-only diagnostic", "heuristic", latency_ms=_ms())

    if llm is None:
        return SafetyVerdict(False, "no judge available (fail closed)", "error",
                             latency_ms=_ms())
    try:
        res = await llm.complete_text(
            JUDGE_PROMPT.format(command=command[:2000]),
            SamplingParams(temperature=0.0, max_tokens=120,
                           grammar=JsonGrammar(soft_close_bytes=80)),
        )
        doc = json.loads(res.text)
        safe = bool(doc.get("safe", False))
        return SafetyVerdict(safe, str(doc.get("reason", "")), "llm", latency_ms=_ms())
    except Exception as e:  # noqa: BLE001 - fail closed on any judge failure
        if org_store is not None:
            org_store.audit("judge_error", str(e)[:500])
        return SafetyVerdict(False, f"judge failure (fail closed): {e}", "error",
                             latency_ms=_ms())
