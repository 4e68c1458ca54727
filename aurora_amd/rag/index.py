"""HBM-resident RAG index with hybrid (vector + BM25) search.

Replaces Weaviate HNSW + text2vec (reference:
routes/knowledge_base/weaviate_client.py:115,224-252 hybrid search) with:
* an on-GPU sentence encoder (models/encoder.py);
* a brute-force kNN over an embedding matrix kept resident in HBM
  (288 GB/GPU holds ~190M 768-dim bf16 chunks; at the reference's scale a
  fused scan beats index maintenance — SURVEY.md §7 phase 2);
* a host-side BM25 inverted index for the lexical half, fused with the
  vector score exactly like Weaviate's hybrid alpha.
"""
from __future__ import annotations

import math
import re
import threading
from collections import Counter, defaultdict
from typing import Dict, List, Optional

import torch

from .. import ops
from ..engine.tokenizer import Tokenizer

_WORD = re.compile(r"[a-z0-9_]+")


def _tokens(text: str) -> List[str]:
    return _WORD.findall(text.lower())


class BM25:
    """Inverted-index BM25: per-term postings make a query cost the sum of
    its terms' posting lists, not O(corpus) per term."""

    def __init__(self, k1: float = 1.2, b: float = 0.75):
        self.k1, self.b = k1, b
        self.postings: Dict[str, Dict[int, int]] = defaultdict(dict)
        self.doc_len: List[int] = []
        self._len_sum = 0

    def add(self, text: str) -> None:
        i = len(self.doc_len)
        terms = Counter(_tokens(text))
        n = sum(terms.values())
        self.doc_len.append(n)
        self._len_sum += n
        for t, f in terms.items():
            self.postings[t][i] = f

    def scores(self, query: str) -> Dict[int, float]:
        q = _tokens(query)
        N = len(self.doc_len)
        if N == 0:
            return {}
        avg = self._len_sum / N
        out: Dict[int, float] = defaultdict(float)
        for term in q:
            plist = self.postings.get(term)
            if not plist:
                continue
            n = len(plist)
            idf = math.log(1 + (N - n + 0.5) / (n + 0.5))
            for i, f in plist.items():
                out[i] += idf * f * (self.k1 + 1) / (
                    f + self.k1 * (1 - self.b + self.b * self.doc_len[i] / avg))
        return out


def chunk_text(text: str, target_chars: int = 1200, overlap: int = 150) -> List[str]:
    """Paragraph-aware chunking (reference: document_processor.py)."""
    paras = [p.strip() for p in text.split("\n\n") if p.strip()]
    chunks: List[str] = []
    cur = ""
    for p in paras:
        if len(cur) + len(p) + 2 <= target_chars:
            cur = (cur + "\n\n" + p).strip()
            continue
        if cur:
            chunks.append(cur)
        while len(p) > target_chars:
            chunks.append(p[:target_chars])
            p = p[target_chars - overlap:]
        cur = p
    if cur:
        chunks.append(cur)
    return chunks or ([text[:target_chars]] if text.strip() else [])


class RagIndex:
    def __init__(self, encoder=None, tokenizer: Optional[Tokenizer] = None,
                 device: str = "cpu", alpha: float = 0.5):
        self.encoder = encoder
        self.tokenizer = tokenizer
        self.device = device
        self.alpha = alpha  # hybrid weight: alpha*vector + (1-alpha)*bm25
        self.chunks: List[dict] = []  # {text, doc_id, doc_name}
        self.emb: Optional[torch.Tensor] = None  # [N, H] on device
        self.bm25 = BM25()
        self._lock = threading.Lock()

    def _embed(self, texts: List[str]) -> Optional[torch.Tensor]:
        if self.encoder is None or self.tokenizer is None:
            return None
        ids = [self.tokenizer.encode(t)[:256] for t in texts]
        return self.encoder.encode_ids(ids).float()

    def add_document(self, doc_id: str, name: str, text: str) -> int:
        parts = chunk_text(text)
        vecs = self._embed(parts)
        with self._lock:
            for i, p in enumerate(parts):
                self.chunks.append({"text": p, "doc_id": doc_id, "doc_name": name})
                self.bm25.add(p)
            if vecs is not None:
                vecs = vecs.to(self.device)
                self.emb = vecs if self.emb is None else torch.cat([self.emb, vecs])
        return len(parts)

    def __len__(self) -> int:
        return len(self.chunks)

    def search(self, query: str, k: int = 5) -> List[dict]:
        from ..utils.streams import on_aux_stream
        with self._lock:
            n = len(self.chunks)
            if n == 0:
                return []
            lex = self.bm25.scores(query)
            lex_max = max(lex.values()) if lex else 1.0
            vec_scores: Dict[int, float] = {}
            if self.emb is not None:
                # aux stream: the encoder + kNN + host read must not queue
                # behind the engine's decode steps (utils/streams.py)
                with on_aux_stream():
                    qv = self._embed([query])
                    if qv is not None:
                        vals, idx = ops.knn_topk(
                            qv.to(self.emb.device).to(self.emb.dtype),
                            self.emb, min(4 * k, n))
                        for v, i in zip(vals[0].tolist(), idx[0].tolist()):
                            vec_scores[i] = v
            cands = set(vec_scores) | set(lex)
            scored = []
            for i in cands:
                s = (self.alpha * vec_scores.get(i, 0.0)
                     + (1 - self.alpha) * lex.get(i, 0.0) / lex_max)
                scored.append((s, i))
            scored.sort(reverse=True)
            return [dict(self.chunks[i], score=s) for s, i in scored[:k]]

    @classmethod
    def from_store(cls, org_store, encoder=None, tokenizer=None,
                   device: str = "cpu") -> "RagIndex":
        """Rebuild the in-memory/HBM index from persisted kb_chunks."""
        idx = cls(encoder, tokenizer, device)
        docs = {d["id"]: d["name"] for d in org_store.kb_documents()}
        by_doc: Dict[str, List[dict]] = defaultdict(list)
        for c in org_store.kb_chunks():
            by_doc[c["doc_id"]].append(c)
        for doc_id, chunks in by_doc.items():
            text = "\n\n".join(c["text"] for c in sorted(chunks, key=lambda c: c["seq"]))
            idx.add_document(doc_id, docs.get(doc_id, doc_id), text)
        return idx
