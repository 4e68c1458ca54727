"""Visualization extraction from RCA transcripts (flag-gated, reference:
chat/background/visualization_generator.py / visualization_extractor.py and
the SSE stream at routes/visualization_stream.py; VISUALIZATION_ENABLED).

Extracts chartable numeric series from tool observations (metric-style
`<label> <value>` and timestamped lines) into chart-spec artifacts and
publishes them on the event bus (the SSE stream carries them)."""
from __future__ import annotations

import json
import re
from typing import Dict, List

_SERIES = re.compile(
    r"^(?P<label>[A-Za-z][\w\-./ %]{2,40}?)[:=\s]\s*(?P<value>-?\d+(?:\.\d+)?)\s*"
    r"(?P<unit>%|ms|s|MiB|GiB|MB|GB|req/s|qps)?\s*$")


def extract_series(messages: List[dict], max_series: int = 6) -> List[dict]:
    series: Dict[str, dict] = {}
    for m in messages:
        if m.get("role") != "observation":
            continue
        for line in (m.get("content") or "").splitlines():
            mt = _SERIES.match(line.strip())
            if not mt:
                continue
            label = mt.group("label").strip()
            s = series.setdefault(label, {"label": label,
                                          "unit": mt.group("unit") or "",
                                          "values": []})
            s["values"].append(float(mt.group("value")))
    out = [s for s in series.values() if len(s["values"]) >= 2]
    out.sort(key=lambda s: -len(s["values"]))
    return out[:max_series]


def build_charts(messages: List[dict]) -> List[dict]:
    return [{
        "type": "line",
        "title": s["label"],
        "unit": s["unit"],
        "points": [{"x": i, "y": v} for i, v in enumerate(s["values"])],
    } for s in extract_series(messages)]


def publish_visualizations(rt, org_id: str, incident_id: str,
                           messages: List[dict]) -> int:
    if not rt.settings.flags.get("visualization", True):
        return 0
    charts = build_charts(messages)
    if charts:
        store = rt.org(org_id)
        store.save_artifact(f"visualizations:{incident_id}",
                            json.dumps(charts, indent=1))
        rt.events.publish(org_id, {"type": "visualization",
                                   "incident_id": incident_id,
                                   "charts": charts})
    return len(charts)
