"""GitHub incident-window correlation + Jenkins build-failure analysis.

github_rca (reference: tools/github_rca_tool.py, 733 LoC): pin the
incident window, pull commits/PRs merged shortly before it, score them
for risk signals, and report the correlation. jenkins_rca (reference:
jenkins_rca tool): recent builds of a job, failure result + console tail
excerpt parsing.
"""
from __future__ import annotations

import json
import re
from typing import Dict, List, Optional

from .base import (fail_closed,
                   ProviderAPIError, Request, Transport, iso, parse_time,
                   truncate_items)

# ------------------------------------------------------------------ GitHub

#: paths whose change shortly before an incident is a risk signal
_RISKY_PATH = re.compile(
    r"(deploy|helm|chart|k8s|kustomize|terraform|\.tf$|dockerfile|"
    r"docker-compose|migration|schema|config|\.env|settings|infra)",
    re.IGNORECASE)
_RISKY_MSG = re.compile(
    r"(revert|hotfix|rollback|force|bump|upgrade|downgrade|migrat|"
    r"feature.flag|config|timeout|pool|limit)", re.IGNORECASE)


def _gh_headers(creds: dict) -> Dict[str, str]:
    if not creds.get("token"):
        raise ProviderAPIError("github credentials need token")
    return {"Authorization": f"Bearer {creds['token']}",
            "Accept": "application/vnd.github+json"}


def build_commits_request(creds: dict, repo: str, incident_time: str,
                          lookback_h: int = 24,
                          now: Optional[float] = None) -> Request:
    t = parse_time(incident_time, now=now)
    return Request("GET", f"https://api.github.com/repos/{repo}/commits",
                   headers=_gh_headers(creds),
                   params={"since": iso(t - lookback_h * 3600),
                           "until": iso(t), "per_page": 50})


def build_commit_detail(creds: dict, repo: str, sha: str) -> Request:
    return Request("GET",
                   f"https://api.github.com/repos/{repo}/commits/{sha}",
                   headers=_gh_headers(creds))


def build_pulls_request(creds: dict, repo: str) -> Request:
    return Request("GET", f"https://api.github.com/repos/{repo}/pulls",
                   headers=_gh_headers(creds),
                   params={"state": "closed", "sort": "updated",
                           "direction": "desc", "per_page": 30})


def score_commit(commit: dict, files: Optional[List[dict]] = None) -> tuple:
    """(score, reasons) — risk signals in message and touched paths."""
    msg = ((commit.get("commit") or {}).get("message") or "").splitlines()[0]
    score = 0
    reasons = []
    if _RISKY_MSG.search(msg):
        score += 2
        reasons.append("risky message")
    for f in files or []:
        if _RISKY_PATH.search(f.get("filename", "")):
            score += 3
            reasons.append(f"touches {f.get('filename', '')[:60]}")
            break
    total = sum((f.get("additions", 0) + f.get("deletions", 0))
                for f in files or [])
    if total > 500:
        score += 1
        reasons.append(f"large change ({total} lines)")
    return score, reasons


def correlate(commits: List[dict], incident_time: str, pulls: List[dict],
              files_by_sha: Optional[Dict[str, List[dict]]] = None,
              now: Optional[float] = None) -> str:
    t_inc = parse_time(incident_time, now=now)
    scored = []
    for c in commits:
        sha = c.get("sha", "")[:10]
        cd = ((c.get("commit") or {}).get("author") or {}).get("date", "")
        try:
            age_min = (t_inc - parse_time(cd, now=now)) / 60.0
        except ProviderAPIError:
            age_min = -1
        s, reasons = score_commit(c, (files_by_sha or {}).get(c.get("sha")))
        if 0 <= age_min <= 120:
            s += 2
            reasons.append(f"{age_min:.0f}min before incident")
        msg = ((c.get("commit") or {}).get("message") or "").splitlines()[0]
        scored.append((s, f"{sha} ({cd}): {msg[:120]}"
                       + (f" [{'; '.join(reasons)}]" if reasons else "")))
    scored.sort(key=lambda x: -x[0])
    merged = [p for p in pulls
              if p.get("merged_at")
              and 0 <= (t_inc - parse_time(p["merged_at"], now=now)) <= 86400]
    lines = [f"commits in the 24h before the incident: {len(commits)}; "
             f"PRs merged: {len(merged)}"]
    lines += ["suspect commits (risk-ranked):"]
    lines += ["  " + row for _, row in scored[:10]] or ["  (none)"]
    if merged:
        lines.append("recently merged PRs:")
        lines += [f"  #{p.get('number')} {p.get('title', '')[:100]} "
                  f"(merged {p.get('merged_at')})" for p in merged[:8]]
    return "\n".join(lines)


async def github_rca(transport: Transport, creds: dict, repo: str,
                     incident_time: str, now: Optional[float] = None) -> str:
    commits = await transport(build_commits_request(creds, repo,
                                                    incident_time, now=now))
    if isinstance(commits, dict):
        raise ProviderAPIError(str(commits.get("message", commits))[:200])
    pulls = await transport(build_pulls_request(creds, repo))
    if isinstance(pulls, dict):
        pulls = []
    # enrich the 5 newest commits with file lists (risk path scoring)
    files_by_sha: Dict[str, List[dict]] = {}
    for c in commits[:5]:
        try:
            det = await transport(build_commit_detail(creds, repo,
                                                      c.get("sha", "")))
            files_by_sha[c.get("sha")] = det.get("files", [])
        except ProviderAPIError:
            continue
    return (f"github_rca for {repo} @ {incident_time}:\n"
            + correlate(commits, incident_time, pulls, files_by_sha, now=now))


# ----------------------------------------------------------------- Jenkins

# ---------------------------------------------------------------- spinnaker


def build_spinnaker_executions(creds: dict, application: str) -> Request:
    """Recent pipeline executions via Gate (reference: spinnaker_rca)."""
    if not creds.get("base_url"):
        raise ProviderAPIError("spinnaker credentials need base_url (gate)")
    h = {}
    if creds.get("token"):
        h["Authorization"] = f"Bearer {creds['token']}"
    return Request("GET", creds["base_url"].rstrip("/")
                   + f"/applications/{application}/pipelines",
                   headers=h, params={"limit": 15, "expand": "false"})


@fail_closed
def parse_spinnaker_executions(doc) -> str:
    if isinstance(doc, dict):
        doc = doc.get("executions", [])
    rows = []
    for e in doc[:15]:
        stages = e.get("stages", [])
        failed = [s.get("name", "?") for s in stages
                  if s.get("status") in ("TERMINAL", "FAILED_CONTINUE")]
        rows.append(f"{e.get('name', '?')} [{e.get('status', '?')}] "
                    f"build {e.get('buildTime', e.get('startTime', ''))}"
                    + (f" failed stages: {', '.join(failed)}" if failed
                       else ""))
    bad = sum(1 for e in doc if e.get("status") not in
              ("SUCCEEDED", "RUNNING", "NOT_STARTED"))
    return (f"spinnaker: {len(doc)} recent executions, {bad} failed"
            + ("\n" + "\n".join(rows) if rows else ""))


async def spinnaker_rca(transport: Transport, creds: dict,
                        application: str, **_kw) -> str:
    return parse_spinnaker_executions(await transport(
        build_spinnaker_executions(creds, application)))


_FAIL_LINE = re.compile(
    r"(error|exception|failed|fatal|traceback|build step.*failed)",
    re.IGNORECASE)


def _jenkins_headers(creds: dict) -> Dict[str, str]:
    if not creds.get("base_url"):
        raise ProviderAPIError("jenkins credentials need base_url")
    h = {}
    if creds.get("user") and creds.get("token"):
        import base64
        h["Authorization"] = "Basic " + base64.b64encode(
            f"{creds['user']}:{creds['token']}".encode()).decode()
    return h


def build_job_builds(creds: dict, job: str) -> Request:
    path = "/".join(f"job/{p}" for p in job.split("/"))
    return Request(
        "GET", creds["base_url"].rstrip("/") + f"/{path}/api/json",
        headers=_jenkins_headers(creds),
        params={"tree": "builds[number,result,timestamp,duration,url]{0,15}"})


def build_console(creds: dict, job: str, number: int) -> Request:
    path = "/".join(f"job/{p}" for p in job.split("/"))
    return Request("GET", creds["base_url"].rstrip("/")
                   + f"/{path}/{number}/consoleText",
                   headers=_jenkins_headers(creds))


@fail_closed
def parse_builds(doc: dict) -> List[dict]:
    return doc.get("builds", [])


def extract_failure_lines(console: str, max_lines: int = 12) -> List[str]:
    """The failing lines of a console log, favoring the tail (reference
    pattern: jenkins_rca reads the last console screens)."""
    lines = console.splitlines()
    hits = [ln.strip()[:200] for ln in lines if _FAIL_LINE.search(ln)]
    return (hits or lines)[-max_lines:]


async def jenkins_rca(transport: Transport, creds: dict, job: str,
                      **_kw) -> str:
    doc = await transport(build_job_builds(creds, job))
    builds = parse_builds(doc)
    if not builds:
        return f"jenkins: job {job!r} has no recorded builds"
    failed = [b for b in builds if b.get("result")
              not in (None, "SUCCESS")][:3]
    lines = [f"jenkins job {job}: last {len(builds)} builds — "
             + ", ".join(f"#{b.get('number')}:{b.get('result') or 'RUNNING'}"
                         for b in builds[:10])]
    for b in failed:
        try:
            console = await transport(build_console(creds, job,
                                                    b.get("number", 0)))
        except ProviderAPIError as e:
            lines.append(f"#{b.get('number')}: console unavailable ({e})")
            continue
        text = console if isinstance(console, str) else json.dumps(console)
        lines.append(f"#{b.get('number')} ({b.get('result')}) failure excerpt:")
        lines += ["  " + ln for ln in extract_failure_lines(text)]
    kept, _ = truncate_items(lines, max_items=60)
    return "\n".join(kept)
