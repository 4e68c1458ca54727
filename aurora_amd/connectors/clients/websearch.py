"""Web search over a self-hosted SearXNG instance.

Rebuilds the reference's WebSearchService (reference:
web_search/web_search_service.py:162-165 + query composer + summarizer,
~2.5k LoC): query composition from incident context, the SearXNG JSON
API request, result parsing/dedup/ranking, and a compact
summary-for-the-agent — all through the pluggable transport so it is
fixture-testable; the live socket is the only n/a-env part.
"""
from __future__ import annotations

import re
from typing import List, Optional

from .base import fail_closed, ProviderAPIError, Request, Transport, truncate_items

#: noise words stripped when composing a search query from alert text
_STOP = {"the", "a", "an", "is", "are", "was", "on", "in", "of", "for",
         "to", "at", "and", "or", "with", "alert", "incident", "error",
         "critical", "warning", "firing", "triggered"}


def compose_query(text: str, extra_terms: Optional[List[str]] = None,
                  max_terms: int = 8) -> str:
    """Alert title/symptom -> search terms: keep identifiers and rare
    words, drop alerting boilerplate (reference: query composer)."""
    toks = re.findall(r"[A-Za-z0-9_.\-/]{3,}", text)
    seen = []
    for t in toks:
        low = t.lower()
        if low in _STOP or low in (s.lower() for s in seen):
            continue
        seen.append(t)
    terms = seen[:max_terms] + list(extra_terms or [])
    return " ".join(terms)


def build_search_request(base_url: str, query: str,
                         categories: str = "it",
                         time_range: str = "") -> Request:
    if not base_url:
        raise ProviderAPIError("no SearXNG base_url configured "
                               "(AURORA_SEARXNG_URL)")
    params = {"q": query, "format": "json", "categories": categories,
              "safesearch": 1}
    if time_range:
        params["time_range"] = time_range  # day | week | month | year
    return Request("GET", base_url.rstrip("/") + "/search", params=params)


@fail_closed
def parse_search_response(doc: dict, max_results: int = 6) -> List[dict]:
    """Dedup by host+title, rank by SearXNG score, strip tracking junk."""
    out: List[dict] = []
    seen = set()
    results = sorted(doc.get("results", []),
                     key=lambda r: -float(r.get("score", 0) or 0))
    for r in results:
        url = str(r.get("url", ""))
        title = str(r.get("title", ""))[:160]
        host = re.sub(r"^https?://([^/]+).*$", r"\1", url)
        key = (host, title.lower())
        if not url or key in seen:
            continue
        seen.add(key)
        out.append({"title": title, "url": url,
                    "content": str(r.get("content", ""))[:400],
                    "engine": r.get("engine", "")})
        if len(out) >= max_results:
            break
    return out


def render_results(results: List[dict], query: str) -> str:
    if not results:
        return f"web search for {query!r}: no results"
    rows = [f"- {r['title']} ({r['url']})\n  {r['content']}"
            for r in results]
    kept, _ = truncate_items(rows, max_items=6, max_chars=4000)
    return f"web search for {query!r}:\n" + "\n".join(kept)


async def web_search(transport: Transport, base_url: str, query: str,
                     time_range: str = "") -> str:
    doc = await transport(build_search_request(base_url, query,
                                               time_range=time_range))
    return render_results(parse_search_response(doc), query)
