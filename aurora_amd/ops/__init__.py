"""Op dispatch layer.

On CUDA (= ROCm/HIP) tensors every op routes to the in-tree gfx950 HIP
extension ``aurora_amd._hip_ops`` and FAILS LOUDLY if it is not built —
there is deliberately no silent eager fallback on the GPU path (the whole
point is the CDNA4 kernels; a quiet PyTorch fallback would fake GPU tests).
On CPU tensors ops route to the fp32 reference (``torch_ref``) so the full
pipeline runs and is testable in CPU-only CI.

Build the extension with ``python setup.py build_ext --inplace`` (hipcc,
``--offload-arch=gfx950``) — see ``aurora_amd/ops/csrc/``.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import torch_ref

_hip = None
_hip_import_error: Optional[BaseException] = None
try:  # the in-tree extension: aurora_amd/_hip_ops*.so
    from aurora_amd import _hip_ops as _hip  # type: ignore
except Exception as e:  # pragma: no cover - exercised only when unbuilt
    _hip_import_error = e


def hip_available() -> bool:
    return _hip is not None


def _require_hip(op: str):
    if _hip is None:
        raise RuntimeError(
            f"aurora_amd HIP extension is not built but op '{op}' was called on a GPU "
            f"tensor. Build it in-tree with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_hip_import_error!r}"
        )
    return _hip


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda:
        return _require_hip("rmsnorm").rmsnorm(x.contiguous(), weight, eps)
    return torch_ref.rmsnorm(x, weight, eps)


def rmsnorm_residual(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
) -> Tuple[torch.Tensor, torch.Tensor]:
    if x.is_cuda:
        m = _require_hip("rmsnorm_residual")
        return m.rmsnorm_residual(x.contiguous(), residual, weight, eps)
    return torch_ref.rmsnorm_residual(x, residual, weight, eps)


def rope_apply(
    q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor, cos_sin: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    if q.is_cuda:
        m = _require_hip("rope_apply")
        q = q.contiguous()
        k = k.contiguous()
        m.rope_apply_(q, k, positions.to(torch.int32), cos_sin)  # in-place
        return q, k
    return torch_ref.rope_apply(q, k, positions, cos_sin)


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    if gate_up.is_cuda:
        return _require_hip("silu_mul").silu_mul(gate_up.contiguous())
    return torch_ref.silu_mul(gate_up)


def attention_prefill(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,
    scale: float,
    causal: bool = True,
) -> torch.Tensor:
    if q.is_cuda:
        m = _require_hip("attention_prefill")
        return m.attention_prefill(
            q.contiguous(), k.contiguous(), v.contiguous(),
            cu_seqlens.to(torch.int32), float(scale), bool(causal),
        )
    return torch_ref.attention_prefill(q, k, v, cu_seqlens, scale, causal)


def attention_prefill_paged(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    cu_seqlens_q: torch.Tensor,
    context_lens: torch.Tensor,
    scale: float,
) -> torch.Tensor:
    if q.is_cuda:
        m = _require_hip("attention_prefill_paged")
        return m.attention_prefill_paged(
            q.contiguous(), k_cache, v_cache, block_tables.to(torch.int32),
            cu_seqlens_q.to(torch.int32), context_lens.to(torch.int32), float(scale),
        )
    return torch_ref.attention_prefill_paged(
        q, k_cache, v_cache, block_tables, cu_seqlens_q, context_lens, scale
    )


def kv_cache_write(
    k: torch.Tensor, v: torch.Tensor,
    k_cache: torch.Tensor, v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    if k.is_cuda:
        m = _require_hip("kv_cache_write")
        m.kv_cache_write(k.contiguous(), v.contiguous(), k_cache, v_cache,
                         slot_mapping.to(torch.int32))
        return
    torch_ref.kv_cache_write(k, v, k_cache, v_cache, slot_mapping)


def paged_attention_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    context_lens: torch.Tensor,
    scale: float,
) -> torch.Tensor:
    if q.is_cuda:
        m = _require_hip("paged_attention_decode")
        return m.paged_attention_decode(
            q.contiguous(), k_cache, v_cache,
            block_tables.to(torch.int32), context_lens.to(torch.int32), float(scale),
        )
    return torch_ref.paged_attention_decode(q, k_cache, v_cache, block_tables, context_lens, scale)


def fused_rope_kv_write(qkv, positions, slots, cos_sin, k_cache, v_cache,
                        n_heads: int, n_kv_heads: int):
    """Split fused qkv output, RoPE q/k, scatter k/v into the paged cache;
    returns contiguous roped q [T, Hq, D].  One kernel on GPU; composed
    reference ops on CPU."""
    if qkv.is_cuda:
        m = _require_hip("fused_rope_kv_write")
        return m.fused_rope_kv_write(qkv.contiguous(), positions.to(torch.int32),
                                     slots.to(torch.int32), cos_sin,
                                     k_cache, v_cache, n_heads, n_kv_heads)
    T = qkv.shape[0]
    D = k_cache.shape[3]
    q = qkv[:, : n_heads * D].reshape(T, n_heads, D)
    k = qkv[:, n_heads * D: (n_heads + n_kv_heads) * D].reshape(T, n_kv_heads, D)
    v = qkv[:, (n_heads + n_kv_heads) * D:].reshape(T, n_kv_heads, D)
    q, k = torch_ref.rope_apply(q, k, positions, cos_sin)
    torch_ref.kv_cache_write(k, v, k_cache, v_cache, slots)
    return q


def embed_pool(hidden: torch.Tensor, cu_seqlens: torch.Tensor) -> torch.Tensor:
    if hidden.is_cuda:
        return _require_hip("embed_pool").embed_pool(hidden.contiguous(), cu_seqlens.to(torch.int32))
    return torch_ref.embed_pool(hidden, cu_seqlens)


def knn_topk(query: torch.Tensor, docs: torch.Tensor, k: int):
    if query.is_cuda:
        return _require_hip("knn_topk").knn_topk(query.contiguous(), docs, int(k))
    return torch_ref.knn_topk(query, docs, k)


def linear(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """Bias-free linear: custom skinny GEMM on the GPU decode path (M<=64),
    hipBLASLt otherwise.  Dispatch thresholds are measured
    (profiles/kernel_perf_r01_*.json): hipBLASLt wins at very large N
    (lm-head shapes), the skinny kernel at mid-size per-layer shapes."""
    import torch.nn.functional as F
    if (x.is_cuda and x.dim() == 2 and 0 < x.shape[0] <= 64
            and x.shape[1] % 128 == 0 and x.dtype == torch.bfloat16
            and _hip is not None):
        w_bytes = weight.shape[0] * weight.shape[1] * 2
        # measured crossover (tools/sweep_skinny2.py, COLD weights rotated
        # past the 256 MB LLC — the in-engine regime): the LDS-staged v2
        # kernel beats hipBLASLt up to ~128 MB of W (o 15.0 vs 21.4 us,
        # down 34.1 vs 42.2 us at M=64); above that (gate_up 235 MB,
        # lm-head) blaslt's large-N kernels reach ~5 TB/s and win.
        import os
        max_mb = int(os.environ.get("AURORA_SKINNY_MAX_MB", "128"))
        if w_bytes <= max_mb * 1024 * 1024:
            s_target = 256 if w_bytes <= 40 * 1024 * 1024 else 512
            return _hip.skinny_gemm2_ex(x.contiguous(), weight, s_target)
    elif (x.is_cuda and x.dim() == 2 and 64 < x.shape[0] <= 256
            and x.shape[1] % 64 == 0 and weight.shape[0] % 64 == 0
            and x.dtype == torch.bfloat16 and _hip is not None):
        # M=65..256 (the B=128-256 decode regime): measured per-shape
        # dispatch (gpurun_out/sweep_skinny3.log, cold rotated weights) —
        # the glds v3 kernel wins the qkv shape at M>128 (1.36 vs 1.18
        # TB/s) and deep-K at M<=128 (down128 2.18 vs 1.59); hipBLASLt
        # keeps o / gate_up / lm-head. Gated off with AURORA_SKINNY3=0.
        import os
        if os.environ.get("AURORA_SKINNY3", "1") != "0":
            M, K, N = x.shape[0], x.shape[1], weight.shape[0]
            if (K >= 8192 and M <= 128) or (4096 < N <= 8192 and K <= 4096
                                            and M > 128):
                return _hip.skinny_gemm3_ex(x.contiguous(), weight, 384)
    return F.linear(x, weight)


def has_moe_grouped(x: torch.Tensor) -> bool:
    """True when the packed MoE grouped-GEMM kernel serves this tensor."""
    import os
    return (x.is_cuda and x.dtype == torch.bfloat16 and _hip is not None
            and x.shape[1] % 64 == 0
            and os.environ.get("AURORA_MOE_GROUPED", "1") != "0")


def moe_grouped_gemm(xp: torch.Tensor, w: torch.Tensor, starts: torch.Tensor,
                     counts: torch.Tensor) -> torch.Tensor:
    """Packed per-expert GEMM: yp[A, N] with tokens sorted by expert.
    Requires per-expert counts <= 256 (the decode regime; prefill uses
    per-expert dense GEMMs)."""
    return _require_hip("moe_grouped_gemm").moe_grouped_gemm(
        xp.contiguous(), w, starts, counts)


def softmax_sample(logits, temperature, top_p, top_k, generator=None):
    # Sampling is [B, V] with B small; keep it in torch on both devices for now
    # (the GPU cost is one softmax over V — negligible next to the forward).
    return torch_ref.softmax_sample(logits, temperature, top_p, top_k, generator)
