"""Sentence-embedding encoder (BGE-class capability, aurora-native arch).

Replaces the reference's Weaviate text2vec-transformers sidecar (reference:
docker-compose.yaml:545) with an in-process encoder running entirely on the
existing gfx950 kernel set: bidirectional varlen attention
(ops.attention_prefill causal=False), RMSNorm, SiLU-MLP, RoPE positions,
mean-pool + L2 (ops.embed_pool).  Weights are random-init here (no network);
the architecture intentionally reuses the serving kernels so the embedder
adds zero new GPU surface.
"""
from __future__ import annotations

from typing import List

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .config import ModelSpec, get_spec


class EncoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, dtype):
        super().__init__()
        h, d = spec.hidden_size, spec.head_dim
        self.n_heads = spec.n_heads
        self.head_dim = d
        self.scale = d ** -0.5
        self.norm1 = nn.Parameter(torch.ones(h, dtype=dtype), requires_grad=False)
        self.norm2 = nn.Parameter(torch.ones(h, dtype=dtype), requires_grad=False)
        self.qkv = nn.Linear(h, 3 * spec.n_heads * d, bias=False, dtype=dtype)
        self.o = nn.Linear(spec.n_heads * d, h, bias=False, dtype=dtype)
        self.gate_up = nn.Linear(h, 2 * spec.intermediate_size, bias=False, dtype=dtype)
        self.down = nn.Linear(spec.intermediate_size, h, bias=False, dtype=dtype)
        for p in self.parameters():
            p.requires_grad_(False)
        self.eps = spec.rms_eps

    def forward(self, x, cos_sin, positions, cu):
        T = x.shape[0]
        h = ops.rmsnorm(x, self.norm1, self.eps)
        qkv = self.qkv(h).view(T, 3 * self.n_heads, self.head_dim)
        q, k, v = qkv.split(self.n_heads, dim=1)
        q, k = ops.rope_apply(q.contiguous(), k.contiguous(), positions, cos_sin)
        o = ops.attention_prefill(q, k, v.contiguous(), cu, self.scale, causal=False)
        x = x + self.o(o.reshape(T, -1))
        h = ops.rmsnorm(x, self.norm2, self.eps)
        return x + self.down(ops.silu_mul(self.gate_up(h)))


class SentenceEncoder(nn.Module):
    def __init__(self, spec_name: str = "bge-base", dtype=torch.bfloat16, device="cpu"):
        super().__init__()
        spec = get_spec(spec_name)
        assert spec.is_encoder, spec_name
        self.spec = spec
        if device == "cpu":
            dtype = torch.float32
        self.embed = nn.Embedding(spec.vocab_size, spec.hidden_size, dtype=dtype)
        self.embed.weight.requires_grad_(False)
        self.layers = nn.ModuleList(
            [EncoderLayer(spec, dtype) for _ in range(spec.n_layers)])
        self.final_norm = nn.Parameter(torch.ones(spec.hidden_size, dtype=dtype),
                                       requires_grad=False)
        cs = ops.torch_ref.rope_cos_sin_cache(spec.head_dim, spec.max_position, 10000.0)
        self.register_buffer("cos_sin", cs, persistent=False)
        self.to(device)
        self.device_str = device

    @torch.inference_mode()
    def init_random(self, seed: int = 0) -> None:
        g = torch.Generator().manual_seed(seed)
        for p in self.parameters():
            if p.dim() >= 2:
                p.copy_((torch.randn(p.shape, generator=g) * p.shape[-1] ** -0.5).to(p.dtype))

    @torch.inference_mode()
    def encode_ids(self, batches: List[List[int]]) -> torch.Tensor:
        """[N, H] L2-normalized embeddings for N token-id lists."""
        dev = self.embed.weight.device
        maxp = self.spec.max_position
        batches = [b[:maxp] if b else [0] for b in batches]
        flat = [t for b in batches for t in b]
        cu = [0]
        pos: List[int] = []
        for b in batches:
            cu.append(cu[-1] + len(b))
            pos.extend(range(len(b)))
        ids = torch.tensor(flat, dtype=torch.long, device=dev)
        cu_t = torch.tensor(cu, dtype=torch.int32, device=dev)
        pos_t = torch.tensor(pos, dtype=torch.int32, device=dev)
        x = self.embed(ids)
        for layer in self.layers:
            x = layer(x, self.cos_sin, pos_t, cu_t)
        x = ops.rmsnorm(x, self.final_norm, self.spec.rms_eps)
        return ops.embed_pool(x, cu_t)
