"""Plain-PyTorch fp32 reference implementations of every hot op.

These are the numerics oracle for the HIP kernels (tests compare the gfx950
kernels against these at fp32) and the CPU execution path for the engine
(so the whole pipeline runs and is testable without a GPU).

Shapes use the packed-varlen convention throughout:
  * prefill: tokens of all scheduled sequences concatenated on dim 0,
    delimited by ``cu_seqlens`` (int32, [num_seqs+1]).
  * decode: one token per sequence, batch on dim 0.

KV cache layout (chosen for the MI355X decode kernel's streaming reads):
  k_cache, v_cache: [num_pages, n_kv_heads, page_size, head_dim]
  slot = page_id * page_size + offset_in_page
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps) * weight.float()
    return y.to(x.dtype)


def rmsnorm_residual(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Fused residual-add + RMSNorm: returns (normed, new_residual)."""
    new_res = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm(new_res, weight, eps), new_res


def rope_cos_sin_cache(
    head_dim: int, max_pos: int, theta: float = 500000.0, device=None, dtype=torch.float32
) -> torch.Tensor:
    """[max_pos, head_dim] cache: first half cos, second half sin (NeoX/llama pairing:
    dims rotate as (i, i + head_dim//2))."""
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(0, half, device=device, dtype=torch.float64) / half))
    t = torch.arange(max_pos, device=device, dtype=torch.float64)
    freqs = torch.outer(t, inv_freq)  # [max_pos, half]
    cache = torch.cat([freqs.cos(), freqs.sin()], dim=-1).to(dtype)
    return cache


def rope_apply(
    q: torch.Tensor,  # [T, n_heads, head_dim]
    k: torch.Tensor,  # [T, n_kv_heads, head_dim]
    positions: torch.Tensor,  # [T] int32/int64
    cos_sin: torch.Tensor,  # [max_pos, head_dim]
) -> Tuple[torch.Tensor, torch.Tensor]:
    half = q.shape[-1] // 2
    cs = cos_sin[positions.long()]  # [T, head_dim]
    cos = cs[:, :half].unsqueeze(1).float()  # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1).float()

    def rot(x: torch.Tensor) -> torch.Tensor:
        xf = x.float()
        x1, x2 = xf[..., :half], xf[..., half:]
        return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1).to(x.dtype)

    return rot(q), rot(k)


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    """[T, 2*I] -> [T, I]: silu(gate) * up, with gate = first half."""
    i = gate_up.shape[-1] // 2
    g = gate_up[..., :i].float()
    u = gate_up[..., i:].float()
    return (g * torch.sigmoid(g) * u).to(gate_up.dtype)


def attention_prefill(
    q: torch.Tensor,  # [T, n_heads, head_dim]
    k: torch.Tensor,  # [T, n_kv_heads, head_dim]
    v: torch.Tensor,  # [T, n_kv_heads, head_dim]
    cu_seqlens: torch.Tensor,  # [num_seqs+1] int32
    scale: float,
    causal: bool = True,
) -> torch.Tensor:
    """Varlen causal attention over freshly-projected K/V (self-contained prefill,
    no cache read). GQA: q heads map to kv head h // (Hq // Hkv)."""
    T, Hq, D = q.shape
    Hkv = k.shape[1]
    g = Hq // Hkv
    out = torch.empty_like(q)
    for s in range(cu_seqlens.numel() - 1):
        a, b = int(cu_seqlens[s]), int(cu_seqlens[s + 1])
        L = b - a
        qs = q[a:b].float()  # [L, Hq, D]
        ks = k[a:b].float().repeat_interleave(g, dim=1)  # [L, Hq, D]
        vs = v[a:b].float().repeat_interleave(g, dim=1)
        att = torch.einsum("qhd,khd->hqk", qs, ks) * scale
        if causal:
            mask = torch.triu(torch.ones(L, L, dtype=torch.bool, device=q.device), diagonal=1)
            att.masked_fill_(mask, float("-inf"))
        p = att.softmax(-1)
        o = torch.einsum("hqk,khd->qhd", p, vs)
        out[a:b] = o.to(q.dtype)
    return out


def attention_prefill_paged(
    q: torch.Tensor,  # [T, n_heads, head_dim] new-chunk queries (packed varlen)
    k_cache: torch.Tensor,  # [num_pages, n_kv_heads, page_size, head_dim]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [num_seqs, max_pages] int32
    cu_seqlens_q: torch.Tensor,  # [num_seqs+1] int32 (query-chunk lengths)
    context_lens: torch.Tensor,  # [num_seqs] int32: total K len incl. the new chunk
    scale: float,
) -> torch.Tensor:
    """Causal attention of a new query chunk against the paged KV cache.

    The new chunk's K/V must already be written to the cache (kv_cache_write).
    Query i (local index within its chunk) has absolute position
    ctx_len - q_len + i and attends to cache slots [0, pos].  Covers both
    fresh prefill (q_len == ctx_len) and append-prefill of a later agent turn
    (q_len < ctx_len) with one op.
    """
    T, Hq, D = q.shape
    Hkv = k_cache.shape[1]
    g = Hq // Hkv
    P = k_cache.shape[2]
    out = torch.empty_like(q)
    for s in range(cu_seqlens_q.numel() - 1):
        a, b = int(cu_seqlens_q[s]), int(cu_seqlens_q[s + 1])
        qlen = b - a
        L = int(context_lens[s])
        npages = (L + P - 1) // P
        pages = block_tables[s, :npages].long()
        ks = k_cache[pages].permute(1, 0, 2, 3).reshape(Hkv, npages * P, D)[:, :L].float()
        vs = v_cache[pages].permute(1, 0, 2, 3).reshape(Hkv, npages * P, D)[:, :L].float()
        qs = q[a:b].float()  # [qlen, Hq, D]
        att = torch.einsum("qhd,hkd->hqk", qs, ks.repeat_interleave(g, dim=0)) * scale
        # causal: query local i (abs pos L - qlen + i) sees keys <= its abs pos
        kpos = torch.arange(L, device=q.device)
        qpos = torch.arange(L - qlen, L, device=q.device)
        mask = kpos.unsqueeze(0) > qpos.unsqueeze(1)  # [qlen, L]
        att.masked_fill_(mask.unsqueeze(0), float("-inf"))
        p = att.softmax(-1)
        o = torch.einsum("hqk,hkd->qhd", p, vs.repeat_interleave(g, dim=0))
        out[a:b] = o.to(q.dtype)
    return out


def kv_cache_write(
    k: torch.Tensor,  # [T, n_kv_heads, head_dim]
    v: torch.Tensor,
    k_cache: torch.Tensor,  # [num_pages, n_kv_heads, page_size, head_dim]
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,  # [T] int32 (page*page_size + offset); -1 = skip
) -> None:
    page_size = k_cache.shape[2]
    valid = slot_mapping >= 0
    slots = slot_mapping[valid].long()
    pages = slots // page_size
    offs = slots % page_size
    k_cache[pages, :, offs] = k[valid].to(k_cache.dtype)
    v_cache[pages, :, offs] = v[valid].to(v_cache.dtype)


def paged_attention_decode(
    q: torch.Tensor,  # [B, n_heads, head_dim] (one new token per seq)
    k_cache: torch.Tensor,  # [num_pages, n_kv_heads, page_size, head_dim]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [B, max_pages_per_seq] int32
    context_lens: torch.Tensor,  # [B] int32 (total tokens incl. the new one)
    scale: float,
) -> torch.Tensor:
    B, Hq, D = q.shape
    Hkv = k_cache.shape[1]
    g = Hq // Hkv
    P = k_cache.shape[2]
    out = torch.empty_like(q)
    for b in range(B):
        L = int(context_lens[b])
        npages = (L + P - 1) // P
        pages = block_tables[b, :npages].long()
        ks = k_cache[pages].permute(1, 0, 2, 3).reshape(Hkv, npages * P, D)[:, :L].float()
        vs = v_cache[pages].permute(1, 0, 2, 3).reshape(Hkv, npages * P, D)[:, :L].float()
        qs = q[b].float()  # [Hq, D]
        att = torch.einsum("hd,hkd->hk", qs, ks.repeat_interleave(g, dim=0)) * scale
        p = att.softmax(-1)
        o = torch.einsum("hk,hkd->hd", p, vs.repeat_interleave(g, dim=0))
        out[b] = o.to(q.dtype)
    return out


def embed_pool(
    hidden: torch.Tensor,  # [T, H]
    cu_seqlens: torch.Tensor,  # [num_seqs+1]
) -> torch.Tensor:
    """Mean-pool per sequence + L2 normalize (BGE/MiniLM-style sentence embedding)."""
    outs = []
    for s in range(cu_seqlens.numel() - 1):
        a, b = int(cu_seqlens[s]), int(cu_seqlens[s + 1])
        m = hidden[a:b].float().mean(0)
        outs.append(m / (m.norm() + 1e-12))
    return torch.stack(outs).to(hidden.dtype)


def knn_topk(
    query: torch.Tensor,  # [Q, D] (normalized)
    docs: torch.Tensor,  # [N, D] (normalized)
    k: int,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Brute-force cosine/IP top-k over an HBM-resident doc matrix."""
    scores = query.float() @ docs.float().t()  # [Q, N]
    k = min(k, docs.shape[0])
    vals, idx = scores.topk(k, dim=-1)
    return vals, idx


def softmax_sample(
    logits: torch.Tensor,  # [B, V] float
    temperature: torch.Tensor,  # [B] (0 => greedy)
    top_p: torch.Tensor,  # [B] in (0, 1]
    top_k: torch.Tensor,  # [B] int (0 => off)
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    """Per-row temperature + top-k + top-p sampling; rows with temperature==0 take argmax."""
    B, V = logits.shape
    out = torch.empty(B, dtype=torch.long, device=logits.device)
    lf = logits.float()
    for b in range(B):
        t = float(temperature[b])
        row = lf[b]
        if t <= 0.0:
            out[b] = row.argmax()
            continue
        row = row / t
        tk = int(top_k[b])
        if tk > 0 and tk < V:
            kth = row.topk(tk).values[-1]
            row = row.masked_fill(row < kth, float("-inf"))
        tp = float(top_p[b])
        if tp < 1.0:
            sorted_row, sorted_idx = row.sort(descending=True)
            probs = sorted_row.softmax(-1)
            cum = probs.cumsum(-1)
            cut = cum - probs >= tp  # tokens strictly after the nucleus
            sorted_row = sorted_row.masked_fill(cut, float("-inf"))
            row = torch.full_like(row, float("-inf")).scatter(0, sorted_idx, sorted_row)
        p = row.softmax(-1)
        out[b] = torch.multinomial(p, 1, generator=generator)[0]
    return out
