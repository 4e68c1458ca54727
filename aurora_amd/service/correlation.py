"""Alert correlation: dedupe/attach incoming alerts to existing incidents.

Rebuilds the reference's AlertCorrelator (reference:
services/correlation/alert_correlator.py:105,363) with the same three
strategies: time-window candidacy, text similarity (on-GPU embeddings when
an encoder is available, lexical Jaccard otherwise — replaces the text2vec
HTTP sidecar of embedding_client.py:20-82), and topology adjacency from the
infrastructure graph.
"""
from __future__ import annotations

import json
import re
import time
from dataclasses import dataclass
from typing import Optional


@dataclass
class CorrelationResult:
    incident_id: Optional[str]  # existing incident to attach to, or None
    score: float
    strategy: str


def _alert_text(title: str, payload: dict) -> str:
    svc = payload.get("service") or payload.get("alert", {}).get("service", "")
    return f"{title} {svc} {payload.get('description', '')}"[:500]


def _jaccard(a: str, b: str) -> float:
    wa = set(re.findall(r"[a-z0-9_\-]{3,}", a.lower()))
    wb = set(re.findall(r"[a-z0-9_\-]{3,}", b.lower()))
    if not wa or not wb:
        return 0.0
    return len(wa & wb) / len(wa | wb)


class AlertCorrelator:
    def __init__(self, org_store, graph=None, rag=None,
                 window_s: float = 900.0, threshold: float = 0.55):
        self.store = org_store
        self.graph = graph
        self.rag = rag  # reuse the RAG encoder for similarity when present
        self.window_s = window_s
        self.threshold = threshold
        self._inc_embs: dict = {}  # incident_id -> cached normalized vec

    def _embed_one(self, text: str):
        """Normalized embedding of one alert text via the shared encoder
        (reference: similarity strategy over text2vec embeddings,
        embedding_client.py:20-82); None when no encoder is attached."""
        if self.rag is None:
            return None
        try:
            v = self.rag._embed([text])
        except Exception:  # noqa: BLE001 - similarity degrades to lexical
            return None
        if v is None:
            return None
        import torch.nn.functional as F
        return F.normalize(v[0].float().cpu(), dim=0)

    def _inc_vec(self, incident_id: str, text: str):
        cached = self._inc_embs.get(incident_id)
        if cached is None:
            cached = self._embed_one(text)
            if cached is not None:
                if len(self._inc_embs) > 512:
                    self._inc_embs.clear()
                self._inc_embs[incident_id] = cached
        return cached

    def correlate(self, title: str, payload: dict,
                  fingerprint: str = "") -> CorrelationResult:
        now = time.time()
        candidates = [i for i in self.store.list_incidents(limit=50)
                      if i["status"] in ("open", "investigating")
                      and now - i["created_at"] <= self.window_s]
        if not candidates:
            return CorrelationResult(None, 0.0, "none")
        # exact fingerprint dedup first
        if fingerprint:
            for alert in self.store.recent_alerts(self.window_s):
                if alert["fingerprint"] == fingerprint and alert["incident_id"]:
                    return CorrelationResult(alert["incident_id"], 1.0, "fingerprint")
        text = _alert_text(title, payload)
        qv = self._embed_one(text)  # [H] or None (no encoder / CPU)
        best: Optional[CorrelationResult] = None
        svc = payload.get("service") or payload.get("alert", {}).get("service")
        for inc in candidates:
            ipay = json.loads(inc.get("payload") or "{}")
            itext = _alert_text(inc["title"], ipay)
            sim = _jaccard(text, itext)
            strategy = "similarity"
            if qv is not None:
                iv = self._inc_vec(inc["id"], itext)
                if iv is not None:
                    cos = float((qv * iv).sum())
                    if cos > sim:
                        sim = cos
                        strategy = "embedding"
            # topology bonus: services adjacent in the dependency graph
            if self.graph is not None and svc:
                isvc = ipay.get("service") or ipay.get("alert", {}).get("service")
                if isvc and isvc != svc:
                    if (svc in self.graph.downstream(isvc)
                            or svc in self.graph.upstream(isvc)):
                        sim += 0.25
                        strategy = "topology"
                elif isvc == svc:
                    sim += 0.3
                    strategy = "same_service"
            if best is None or sim > best.score:
                best = CorrelationResult(inc["id"], sim, strategy)
        if best is not None and best.score >= self.threshold:
            return best
        return CorrelationResult(None, best.score if best else 0.0, "below_threshold")

    def handle_alert(self, source: str, title: str, payload: dict,
                     fingerprint: str = "",
                     user_id: Optional[str] = None) -> tuple[str, bool]:
        """Store the alert; attach to a correlated incident or create a new
        one.  Returns (incident_id, created_new)."""
        res = self.correlate(title, payload, fingerprint)
        if res.incident_id is not None:
            self.store.add_alert(source, payload, fingerprint, res.incident_id)
            self.store.lifecycle(res.incident_id, "alert_correlated",
                                 {"strategy": res.strategy, "score": res.score})
            self.store.touch_incident(res.incident_id)
            # queue a context update for a RUNNING investigation (reference:
            # chat/background/context_updates.py:145,343)
            inc = self.store.get_incident(res.incident_id)
            sid = (inc or {}).get("aurora_chat_session_id")
            if sid:
                row = self.store.get_artifact(f"context_updates:{sid}")
                updates = json.loads(row["content"]) if row else []
                updates.append({"source": source, "title": title,
                                "ts": time.time()})
                self.store.save_artifact(f"context_updates:{sid}",
                                         json.dumps(updates[-10:]))
            return res.incident_id, False
        iid = self.store.create_incident(
            title=title, source=source, payload=payload, user_id=user_id,
            severity=payload.get("severity")
            or payload.get("alert", {}).get("severity"))
        self.store.add_alert(source, payload, fingerprint, iid)
        self.store.lifecycle(iid, "incident_created", {"source": source})
        return iid, True
