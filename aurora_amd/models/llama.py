"""Llama-3 / Qwen2-family decoder (GQA, RoPE, SiLU-MLP, optional QKV bias)
built on the aurora_amd op set.

Every hot op routes through aurora_amd.ops (HIP/CDNA4 kernels on GPU, fp32
torch reference on CPU); plain projections stay on hipBLASLt via F.linear.
The forward is engine-shaped: packed varlen tokens + paged KV metadata, no
[B, S] padding anywhere.

Replaces the reference's rented chat models (reference:
server/chat/backend/agent/llm.py:39-67, providers/) with an in-process
model the engine steps directly.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..parallel import dist as pdist
from ..parallel.tp import (
    ColumnParallelLinear,
    MergedColumnParallelLinear,
    QKVParallelLinear,
    RowParallelLinear,
)
from .config import ModelSpec


@dataclass
class AttnMetadata:
    """Per-step attention metadata shared by all layers.

    mode "prefill": q tokens are packed varlen chunks (cu_seqlens_q); their
    K/V have already been assigned cache slots via slot_mapping and the
    kernel attends against the paged cache (covers fresh and append prefill).
    mode "decode": one q token per sequence.
    """

    mode: str  # "prefill" | "decode" | "mixed"
    positions: torch.Tensor  # [T] int32 absolute positions
    slot_mapping: torch.Tensor  # [T] int32 cache slots for the new K/V
    block_tables: torch.Tensor  # [num_seqs, max_pages] int32
    context_lens: torch.Tensor  # [num_seqs] int32 (incl. new tokens)
    cu_seqlens_q: Optional[torch.Tensor] = None  # [num_seqs+1] int32, prefill only
    # mixed mode: tokens are [prefill chunks..., decode tokens]; the fields
    # above describe the prefill part, the ones below the decode part
    num_prefill_tokens: int = 0
    dec_block_tables: Optional[torch.Tensor] = None
    dec_context_lens: Optional[torch.Tensor] = None


class LlamaAttention(nn.Module):
    def __init__(self, spec: ModelSpec, dtype):
        super().__init__()
        self.spec = spec
        tp = pdist.tp_world_size()
        self.n_heads = spec.n_heads // tp
        self.n_kv_heads = spec.n_kv_heads // tp
        self.head_dim = spec.head_dim
        self.scale = spec.head_dim ** -0.5
        self.qkv_proj = QKVParallelLinear(
            spec.hidden_size, spec.n_heads, spec.n_kv_heads, spec.head_dim,
            dtype=dtype, bias=spec.qkv_bias
        )
        self.o_proj = RowParallelLinear(spec.q_size, spec.hidden_size, dtype=dtype)

    def forward(
        self,
        x: torch.Tensor,  # [T, H]
        cos_sin: torch.Tensor,
        kv_cache: Tuple[torch.Tensor, torch.Tensor],
        meta: AttnMetadata,
    ) -> torch.Tensor:
        T = x.shape[0]
        qkv = self.qkv_proj(x)
        k_cache, v_cache = kv_cache
        # fused split + RoPE + paged K/V scatter (one kernel on GPU)
        q = ops.fused_rope_kv_write(qkv, meta.positions, meta.slot_mapping,
                                    cos_sin, k_cache, v_cache,
                                    self.n_heads, self.n_kv_heads)
        if meta.mode == "prefill":
            o = ops.attention_prefill_paged(
                q, k_cache, v_cache, meta.block_tables,
                meta.cu_seqlens_q, meta.context_lens, self.scale,
            )
        elif meta.mode == "decode":
            o = ops.paged_attention_decode(
                q, k_cache, v_cache, meta.block_tables, meta.context_lens, self.scale
            )
        else:  # mixed: prefill chunks + decode tokens in one forward — the
            # projections/norms run on the whole packed batch (bigger GEMMs),
            # attention splits per regime
            Tp = meta.num_prefill_tokens
            o_pre = ops.attention_prefill_paged(
                q[:Tp], k_cache, v_cache, meta.block_tables,
                meta.cu_seqlens_q, meta.context_lens, self.scale,
            ) if Tp else q[:0]
            o_dec = ops.paged_attention_decode(
                q[Tp:].contiguous(), k_cache, v_cache, meta.dec_block_tables,
                meta.dec_context_lens, self.scale,
            ) if T > Tp else q[:0]
            o = torch.cat([o_pre, o_dec]) if (Tp and T > Tp) else (o_pre if Tp else o_dec)
        return self.o_proj(o.reshape(T, self.n_heads * self.head_dim))


class LlamaMLP(nn.Module):
    def __init__(self, spec: ModelSpec, dtype):
        super().__init__()
        self.gate_up = MergedColumnParallelLinear(spec.hidden_size, spec.intermediate_size, dtype=dtype)
        self.down = RowParallelLinear(spec.intermediate_size, spec.hidden_size, dtype=dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down(ops.silu_mul(self.gate_up(x)))


class _FullWidthMLP(nn.Module):
    """Unsharded SiLU MLP for expert-parallel mode: the whole expert lives
    on its owner rank (no intra-expert TP, no per-expert all-reduce)."""

    def __init__(self, spec: ModelSpec, dtype):
        super().__init__()
        self.gate_up = nn.Parameter(torch.empty(
            2 * spec.intermediate_size, spec.hidden_size, dtype=dtype),
            requires_grad=False)
        self.down = nn.Parameter(torch.empty(
            spec.hidden_size, spec.intermediate_size, dtype=dtype),
            requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.linear(ops.silu_mul(ops.linear(x, self.gate_up)), self.down)


class MoEMLP(nn.Module):
    """Mixtral-style sparse MoE block (top-k routing over dense experts).

    Experts are TP-sharded like the dense MLP (each expert's gate_up column-
    sharded, down row-sharded) so the expert compute stays local and the
    row-parallel all-reduce doubles as the TP combine.  Token->expert
    dispatch is a host-side gather (grouped GEMM fusion is a later round).
    """

    def __init__(self, spec: ModelSpec, dtype):
        super().__init__()
        from ..parallel import ep as pep
        self.n_experts = spec.n_experts
        self.top_k = spec.experts_per_token
        self.router = nn.Linear(spec.hidden_size, spec.n_experts, bias=False, dtype=dtype)
        self.router.weight.requires_grad_(False)
        self.ep_size = pep.ep_size()
        if self.ep_size > 1:
            # expert parallelism (AURORA_EP=1): this rank owns
            # n_experts/ep FULL-width experts; tokens travel to their
            # experts over all-to-all instead of sharding each expert
            assert spec.n_experts % self.ep_size == 0
            self.e_local = spec.n_experts // self.ep_size
            self.experts = nn.ModuleList(
                [_FullWidthMLP(spec, dtype) for _ in range(self.e_local)])
        else:
            self.experts = nn.ModuleList(
                [LlamaMLP(spec, dtype) for _ in range(spec.n_experts)]
            )
        self._w_gu = None  # lazy stacked-weight cache for the batched path
        self._w_dn = None

    def _local_expert_fn(self, xp: torch.Tensor,
                         local_sel: torch.Tensor) -> torch.Tensor:
        y = torch.empty_like(xp)
        for le in range(self.e_local):
            m = (local_sel == le)
            if m.any():
                y[m] = self.experts[le](xp[m])
        return y

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T, H = x.shape
        logits = self.router(x).float()  # [T, E]
        weights, sel = logits.topk(self.top_k, dim=-1)
        weights = weights.softmax(-1).to(x.dtype)  # [T, k]
        if self.ep_size > 1:
            from ..parallel import ep as pep
            return pep.ep_moe_forward(x, weights, sel, self.n_experts,
                                      self._local_expert_fn)
        if T <= 256:
            return self._forward_batched(x, weights, sel)
        # prefill: per-expert token batches are large, so plain per-expert
        # GEMMs are efficient and avoid the capacity buffer
        out = torch.zeros_like(x)
        for e in range(self.n_experts):
            tok, slot = (sel == e).nonzero(as_tuple=True)
            if tok.numel() == 0:
                continue
            out.index_add_(0, tok, self.experts[e](x[tok]) * weights[tok, slot, None])
        return out

    def _forward_batched(self, x: torch.Tensor, weights: torch.Tensor,
                         sel: torch.Tensor) -> torch.Tensor:
        """Decode path: tokens sorted by expert into a PACKED buffer
        (no capacity padding), then per-expert grouped GEMMs via the
        streaming moe_grouped_gemm kernel — unrouted experts skip their
        whole weight stream, activations/silu run over A = T*top_k rows
        instead of E*C (docs/ROUND2.md §4). Static shapes + device-read
        counts keep the step hipGraph-capturable. CPU / no-extension
        fallback: the round-1 fixed-capacity bmm dispatch."""
        T, H = x.shape
        E, k = self.n_experts, self.top_k
        flat = sel.reshape(-1)  # [T*k]
        order = flat.argsort(stable=True)  # tokens grouped by expert
        tok_of = order // k  # token index of each sorted assignment
        # histogram via scatter_add (bincount host-syncs -> not capturable)
        counts = torch.zeros(E, dtype=torch.long, device=x.device).scatter_add_(
            0, flat, torch.ones_like(flat))
        starts = torch.cumsum(counts, 0) - counts
        if self._w_gu is None:  # stacked expert weights, cached (constant)
            self._w_gu = torch.stack([ex.gate_up.weight for ex in self.experts])
            self._w_dn = torch.stack([ex.down.weight for ex in self.experts])
        w_gu, w_dn = self._w_gu, self._w_dn
        wflat = weights.reshape(-1)[order].unsqueeze(1).to(x.dtype)
        if ops.has_moe_grouped(x):
            xp = x[tok_of]  # [A, H] packed, expert-sorted
            st32 = starts.to(torch.int32)
            ct32 = counts.to(torch.int32)
            gu = ops.moe_grouped_gemm(xp, w_gu, st32, ct32)     # [A, 2I']
            act = ops.silu_mul(gu)                              # [A, I']
            ye = ops.moe_grouped_gemm(act, w_dn, st32, ct32)    # [A, H]
            out = torch.zeros_like(x)
            out.index_add_(0, tok_of, ye * wflat)
        else:
            C = T
            rank = torch.arange(T * k, device=x.device) - starts[flat[order]]
            slot_idx = flat[order] * C + rank  # rank < C always
            xe = x.new_zeros(E * C, H)
            xe[slot_idx] = x[tok_of]
            xe = xe.view(E, C, H)
            gu = torch.bmm(xe, w_gu.transpose(1, 2))  # [E, C, 2I']
            act = ops.silu_mul(gu.reshape(E * C, -1)).view(E, C, -1)
            ye = torch.bmm(act, w_dn.transpose(1, 2)).view(E * C, H)
            out = torch.zeros_like(x)
            out.index_add_(0, tok_of, ye[slot_idx] * wflat)
        from ..parallel import dist as pdist
        return pdist.tp_all_reduce(out) if pdist.tp_world_size() > 1 else out


class LlamaDecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, dtype):
        super().__init__()
        self.input_norm = nn.Parameter(torch.empty(spec.hidden_size, dtype=dtype), requires_grad=False)
        self.post_attn_norm = nn.Parameter(torch.empty(spec.hidden_size, dtype=dtype), requires_grad=False)
        self.eps = spec.rms_eps
        self.attn = LlamaAttention(spec, dtype)
        self.mlp = MoEMLP(spec, dtype) if spec.n_experts > 0 else LlamaMLP(spec, dtype)

    def forward(self, x, residual, cos_sin, kv_cache, meta):
        if residual is None:
            residual = x
            h = ops.rmsnorm(x, self.input_norm, self.eps)
        else:
            h, residual = ops.rmsnorm_residual(x, residual, self.input_norm, self.eps)
        a = self.attn(h, cos_sin, kv_cache, meta)
        h, residual = ops.rmsnorm_residual(a, residual, self.post_attn_norm, self.eps)
        m = self.mlp(h)
        return m, residual


class LlamaForCausalLM(nn.Module):
    def __init__(self, spec: ModelSpec, dtype=torch.bfloat16, device=None):
        super().__init__()
        self.spec = spec
        self.dtype = dtype
        self.embed = nn.Embedding(spec.vocab_size, spec.hidden_size, dtype=dtype)
        self.embed.weight.requires_grad_(False)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(spec, dtype) for _ in range(spec.n_layers)]
        )
        self.final_norm = nn.Parameter(torch.empty(spec.hidden_size, dtype=dtype), requires_grad=False)
        if spec.tie_embeddings:
            self.lm_head = None
        else:
            self.lm_head = nn.Linear(spec.hidden_size, spec.vocab_size, bias=False, dtype=dtype)
            self.lm_head.weight.requires_grad_(False)
        cos_sin = ops.torch_ref.rope_cos_sin_cache(
            spec.head_dim, spec.max_position, spec.rope_theta, dtype=torch.float32
        )
        self.register_buffer("cos_sin", cos_sin, persistent=False)
        if device is not None:
            self.to(device)

    @torch.inference_mode()
    def forward(
        self,
        input_ids: torch.Tensor,  # [T]
        kv_caches: List[Tuple[torch.Tensor, torch.Tensor]],
        meta: AttnMetadata,
    ) -> torch.Tensor:
        x = self.embed(input_ids)
        residual = None
        for layer, cache in zip(self.layers, kv_caches):
            x, residual = layer(x, residual, self.cos_sin, cache, meta)
        x = ops.rmsnorm_residual(x, residual, self.final_norm, self.spec.rms_eps)[0]
        return x  # hidden states [T, H]

    @torch.inference_mode()
    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        w = self.embed.weight if self.lm_head is None else self.lm_head.weight
        return ops.linear(hidden, w).float()

    @torch.inference_mode()
    def init_random(self, seed: int = 0) -> None:
        """Random-init all weights (synthetic-bench environment: no checkpoints).

        Small params use a CPU generator so CPU and GPU instances of the same
        spec get identical weights (engine equivalence tests); big params
        (>16M elements, e.g. the 8B model's) are generated on-device for
        speed — cross-device bit-equality is irrelevant at that scale.
        """
        gen = torch.Generator(device="cpu").manual_seed(seed)
        dev_gen = None
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                std = p.shape[-1] ** -0.5
                if p.numel() > (1 << 24) and p.device.type != "cpu":
                    if dev_gen is None:
                        dev_gen = torch.Generator(device=p.device).manual_seed(seed)
                    r = torch.randn(p.shape, generator=dev_gen, device=p.device,
                                    dtype=torch.float32)
                else:
                    r = torch.randn(p.shape, generator=gen, dtype=torch.float32)
                p.copy_(r.mul_(std).to(p.dtype))
            else:
                p.copy_(torch.ones(p.shape, dtype=p.dtype))
