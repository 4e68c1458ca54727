"""Tensor-parallel linear layers (Megatron-style column/row split).

At tp_world_size()==1 these are plain bias-free linears; with TP the column
layer shards output features (heads / MLP intermediate) and the row layer
shards input features and all-reduces its partial outputs over RCCL/xGMI.
Weights are created pre-sharded (random init in this environment); the
loader shards checkpoints the same way.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from . import dist as pdist


class ColumnParallelLinear(nn.Module):
    """Y_local = X @ W_local^T, W_local = W[rank-th output shard]."""

    def __init__(self, in_features: int, out_features: int, dtype=None,
                 gather_output: bool = False, bias: bool = False):
        super().__init__()
        tp = pdist.tp_world_size()
        assert out_features % tp == 0, (out_features, tp)
        self.in_features = in_features
        self.out_features = out_features
        self.out_local = out_features // tp
        self.gather_output = gather_output
        self.weight = nn.Parameter(
            torch.empty(self.out_local, in_features, dtype=dtype), requires_grad=False
        )
        # output-feature bias shards with the weight rows (Qwen2-family
        # attention projections carry a bias; Llama/Mixtral do not)
        self.bias = nn.Parameter(
            torch.empty(self.out_local, dtype=dtype), requires_grad=False
        ) if bias else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = ops.linear(x, self.weight)
        if self.bias is not None:
            y = y + self.bias
        if self.gather_output:
            y = pdist.tp_all_gather(y, dim=-1)
        return y


class RowParallelLinear(nn.Module):
    """Y = all_reduce(X_local @ W_local^T), W_local = W[rank-th input shard]."""

    def __init__(self, in_features: int, out_features: int, dtype=None):
        super().__init__()
        tp = pdist.tp_world_size()
        assert in_features % tp == 0, (in_features, tp)
        self.in_features = in_features
        self.in_local = in_features // tp
        self.out_features = out_features
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_local, dtype=dtype), requires_grad=False
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = ops.linear(x, self.weight)
        return pdist.tp_all_reduce(y)


class QKVParallelLinear(ColumnParallelLinear):
    """Fused QKV projection, head-sharded: local output is
    [q_heads_local*D | kv_heads_local*D | kv_heads_local*D]."""

    def __init__(self, hidden: int, n_heads: int, n_kv_heads: int, head_dim: int,
                 dtype=None, bias: bool = False):
        tp = pdist.tp_world_size()
        assert n_heads % tp == 0 and n_kv_heads % tp == 0, (n_heads, n_kv_heads, tp)
        self.n_heads_local = n_heads // tp
        self.n_kv_heads_local = n_kv_heads // tp
        self.head_dim = head_dim
        out = (n_heads + 2 * n_kv_heads) * head_dim
        super().__init__(hidden, out, dtype=dtype, bias=bias)

    def split_qkv(self, qkv: torch.Tensor):
        d = self.head_dim
        q = qkv[..., : self.n_heads_local * d]
        k = qkv[..., self.n_heads_local * d: (self.n_heads_local + self.n_kv_heads_local) * d]
        v = qkv[..., (self.n_heads_local + self.n_kv_heads_local) * d:]
        return q, k, v


class MergedColumnParallelLinear(ColumnParallelLinear):
    """gate|up fused projection: each half is sharded independently so the
    local output stays [gate_local | up_local] (silu_mul expects that)."""

    def __init__(self, hidden: int, intermediate: int, dtype=None):
        super().__init__(hidden, 2 * intermediate, dtype=dtype)
        self.intermediate_local = intermediate // pdist.tp_world_size()
