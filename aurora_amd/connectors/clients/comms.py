"""Slack, Jira, Confluence and Notion clients (reference: the comms/docs
tool cluster — slack channel/history/search, jira JQL search with
pagination, confluence CQL search + HTML stripping, notion workspace
search)."""
from __future__ import annotations

import json
import re
from typing import List, Optional

from .base import (fail_closed,
                   ProviderAPIError, Request, Transport, truncate_items,
                   window)

# ------------------------------------------------------------------ Slack


def build_slack_search(creds: dict, query: str) -> Request:
    if not creds.get("token"):
        raise ProviderAPIError("slack credentials need token")
    return Request("GET", "https://slack.com/api/search.messages",
                   headers={"Authorization": f"Bearer {creds['token']}"},
                   params={"query": query, "count": 20, "sort": "timestamp"})


def build_slack_history(creds: dict, channel: str,
                        time_from: Optional[str] = None,
                        time_to: Optional[str] = None,
                        now: Optional[float] = None) -> Request:
    f, t = window(time_from, time_to, default_span_s=7200, now=now)
    return Request("GET", "https://slack.com/api/conversations.history",
                   headers={"Authorization": f"Bearer {creds['token']}"},
                   params={"channel": channel, "oldest": f"{f:.6f}",
                           "latest": f"{t:.6f}", "limit": 50})


def build_slack_channels(creds: dict) -> Request:
    return Request("GET", "https://slack.com/api/conversations.list",
                   headers={"Authorization": f"Bearer {creds['token']}"},
                   params={"exclude_archived": "true", "limit": 200,
                           "types": "public_channel"})


def _slack_ok(doc: dict) -> dict:
    if not doc.get("ok", False):
        raise ProviderAPIError(f"slack API error: {doc.get('error', '?')}")
    return doc


@fail_closed
def parse_slack_search(doc: dict) -> str:
    _slack_ok(doc)
    matches = doc.get("messages", {}).get("matches", [])
    rows = [f"#{m.get('channel', {}).get('name', '?')} "
            f"@{m.get('username', '?')} [{m.get('ts', '')}]: "
            f"{(m.get('text') or '')[:240]}" for m in matches[:20]]
    return (f"slack search: {doc.get('messages', {}).get('total', len(rows))} matches"
            + ("\n" + "\n".join(rows) if rows else " — none"))


@fail_closed
def parse_slack_history(doc: dict) -> str:
    _slack_ok(doc)
    rows = [f"[{m.get('ts', '')}] {m.get('user', m.get('bot_id', '?'))}: "
            f"{(m.get('text') or '')[:240]}"
            for m in doc.get("messages", [])[:50]]
    return ("slack history:" + ("\n" + "\n".join(rows) if rows else " empty"))


@fail_closed
def parse_slack_channels(doc: dict) -> str:
    _slack_ok(doc)
    rows = [f"#{c.get('name')} ({c.get('id')}, members={c.get('num_members', '?')})"
            for c in doc.get("channels", [])[:100]]
    return "slack channels:\n" + "\n".join(rows) if rows else "no channels"


async def slack_query(transport: Transport, creds: dict, query: str,
                      **_kw) -> str:
    q = (query or "").strip()
    if q.lower() in ("channels", "list channels"):
        return parse_slack_channels(await transport(build_slack_channels(creds)))
    if q.lower().startswith("history "):
        return parse_slack_history(await transport(
            build_slack_history(creds, q.split(None, 1)[1].strip())))
    return parse_slack_search(await transport(build_slack_search(creds, q)))


# ------------------------------------------------------------------- Jira

JIRA_FIELDS = ["key", "summary", "status", "priority", "assignee",
               "updated", "resolution"]


def build_jira_search(creds: dict, jql: str, start_at: int = 0,
                      max_results: int = 25) -> Request:
    if not creds.get("base_url") or not creds.get("token"):
        raise ProviderAPIError("jira credentials need base_url + token")
    headers = {"Content-Type": "application/json"}
    if creds.get("email"):  # cloud basic auth pair
        import base64
        tok = base64.b64encode(
            f"{creds['email']}:{creds['token']}".encode()).decode()
        headers["Authorization"] = f"Basic {tok}"
    else:
        headers["Authorization"] = f"Bearer {creds['token']}"
    return Request("POST", creds["base_url"].rstrip("/") + "/rest/api/2/search",
                   headers=headers,
                   json_body={"jql": jql, "startAt": start_at,
                              "maxResults": max_results,
                              "fields": JIRA_FIELDS})


@fail_closed
def parse_jira_search(doc: dict) -> tuple:
    """-> (rows, total)."""
    rows: List[str] = []
    for i in doc.get("issues", []):
        f = i.get("fields", {})
        rows.append(
            f"{i.get('key')} [{(f.get('status') or {}).get('name', '?')}/"
            f"{(f.get('priority') or {}).get('name', '-')}] "
            f"{(f.get('summary') or '')[:140]} "
            f"(assignee={((f.get('assignee') or {}).get('displayName')) or 'unassigned'}, "
            f"updated={f.get('updated', '')})")
    return rows, int(doc.get("total", len(rows)))


async def jira_query(transport: Transport, creds: dict, jql: str,
                     **_kw) -> str:
    rows: List[str] = []
    start = 0
    total = None
    for _ in range(3):  # startAt pagination
        doc = await transport(build_jira_search(creds, jql, start_at=start))
        page, total = parse_jira_search(doc)
        rows.extend(page)
        start += len(page)
        if not page or start >= total:
            break
    kept, trunc = truncate_items(rows, max_items=40)
    head = f"jira: {total} issues for {jql!r}" + (" (truncated)" if trunc else "")
    return head + ("\n" + "\n".join(kept) if kept else " — none")


# -------------------------------------------------------------- Confluence

_TAG = re.compile(r"<[^>]+>")


def strip_html(s: str) -> str:
    return re.sub(r"\s+", " ", _TAG.sub(" ", s or "")).strip()


def build_confluence_search(creds: dict, text: str,
                            limit: int = 10) -> Request:
    if not creds.get("base_url") or not creds.get("token"):
        raise ProviderAPIError("confluence credentials need base_url + token")
    cql = f'text ~ "{text.replace(chr(34), "")}" order by lastmodified desc'
    return Request("GET",
                   creds["base_url"].rstrip("/") + "/rest/api/content/search",
                   headers={"Authorization": f"Bearer {creds['token']}"},
                   params={"cql": cql, "limit": limit,
                           "expand": "body.view,space,version"})


@fail_closed
def parse_confluence_search(doc: dict) -> str:
    rows = []
    for p in doc.get("results", [])[:10]:
        body = strip_html(((p.get("body") or {}).get("view") or {})
                          .get("value", ""))[:500]
        rows.append(f"[{(p.get('space') or {}).get('key', '?')}] "
                    f"{p.get('title', '?')}: {body}")
    return (f"confluence: {doc.get('size', len(rows))} pages"
            + ("\n" + "\n".join(rows) if rows else " — none match"))


async def confluence_query(transport: Transport, creds: dict, text: str,
                           **_kw) -> str:
    return parse_confluence_search(await transport(
        build_confluence_search(creds, text)))


# ----------------------------------------------------------------- Notion


def build_notion_search(creds: dict, text: str) -> Request:
    if not creds.get("token"):
        raise ProviderAPIError("notion credentials need token")
    return Request("POST", "https://api.notion.com/v1/search",
                   headers={"Authorization": f"Bearer {creds['token']}",
                            "Notion-Version": "2022-06-28",
                            "Content-Type": "application/json"},
                   json_body={"query": text, "page_size": 15,
                              "sort": {"direction": "descending",
                                       "timestamp": "last_edited_time"}})


def _notion_title(page: dict) -> str:
    props = page.get("properties", {})
    for p in props.values():
        if p.get("type") == "title":
            return "".join(t.get("plain_text", "")
                           for t in p.get("title", []))[:140]
    return page.get("id", "?")


def build_notion_postmortem(creds: dict, parent_page_id: str, title: str,
                            markdown: str) -> Request:
    """Publish a postmortem as a Notion page (reference: notion postmortem
    suite ~2.6k LoC): markdown -> Notion block objects (headings, bullets,
    paragraphs — the shapes build_postmortem emits)."""
    if not creds.get("token"):
        raise ProviderAPIError("notion credentials need token")
    blocks: List[dict] = []
    for raw in markdown.splitlines():
        line = raw.rstrip()
        if not line.strip():
            continue
        if line.startswith("### "):
            t, key = line[4:], "heading_3"
        elif line.startswith("## "):
            t, key = line[3:], "heading_2"
        elif line.startswith("# "):
            t, key = line[2:], "heading_1"
        elif line.lstrip().startswith("- "):
            t, key = line.lstrip()[2:], "bulleted_list_item"
        else:
            t, key = line, "paragraph"
        blocks.append({"object": "block", "type": key,
                       key: {"rich_text": [{"type": "text",
                                            "text": {"content": t[:1990]}}]}})
        if len(blocks) >= 95:  # Notion caps children per request at 100
            break
    return Request("POST", "https://api.notion.com/v1/pages",
                   headers={"Authorization": f"Bearer {creds['token']}",
                            "Notion-Version": "2022-06-28",
                            "Content-Type": "application/json"},
                   json_body={"parent": {"page_id": parent_page_id},
                              "properties": {"title": {"title": [
                                  {"type": "text",
                                   "text": {"content": title[:200]}}]}},
                              "children": blocks})


async def notion_publish_postmortem(transport: Transport, creds: dict,
                                    parent_page_id: str, title: str,
                                    markdown: str) -> str:
    doc = await transport(build_notion_postmortem(creds, parent_page_id,
                                                  title, markdown))
    if doc.get("object") == "error":
        raise ProviderAPIError(str(doc.get("message", "notion error"))[:200])
    return f"postmortem published to Notion: {doc.get('url', doc.get('id', '?'))}"


# -------------------------------------------------------------- SharePoint


def build_sharepoint_search(creds: dict, text: str) -> Request:
    """Microsoft Graph search over SharePoint/OneDrive content
    (reference: sharepoint search tool)."""
    if not creds.get("token"):
        raise ProviderAPIError("sharepoint credentials need token")
    return Request("POST", "https://graph.microsoft.com/v1.0/search/query",
                   headers={"Authorization": f"Bearer {creds['token']}",
                            "Content-Type": "application/json"},
                   json_body={"requests": [{
                       "entityTypes": ["driveItem", "listItem"],
                       "query": {"queryString": text}, "size": 10}]})


@fail_closed
def parse_sharepoint_search(doc: dict) -> str:
    rows = []
    for v in doc.get("value", []):
        for c in v.get("hitsContainers", []):
            for h in c.get("hits", [])[:10]:
                res = h.get("resource", {})
                rows.append(f"{res.get('name', res.get('id', '?'))}: "
                            f"{(h.get('summary') or '')[:200]}")
    return ("sharepoint search:" + ("\n" + "\n".join(rows)
                                    if rows else " no results"))


async def sharepoint_query(transport: Transport, creds: dict, text: str,
                           **_kw) -> str:
    return parse_sharepoint_search(await transport(
        build_sharepoint_search(creds, text)))


@fail_closed
def parse_notion_search(doc: dict) -> str:
    rows = [f"[{r.get('object', '?')}] {_notion_title(r)} "
            f"(edited {r.get('last_edited_time', '')})"
            for r in doc.get("results", [])[:15]]
    return ("notion search:"
            + ("\n" + "\n".join(rows) if rows else " no results"))


async def notion_query(transport: Transport, creds: dict, text: str,
                       **_kw) -> str:
    return parse_notion_search(await transport(build_notion_search(creds, text)))
