"""Checkpoint loading: HF-layout safetensors -> the fused/TP-sharded model.

This environment has no network, so every run is random-init — but the
production path a reference user needs is real weights.  Maps HuggingFace
Llama/Qwen2/Mixtral tensor names onto this model's fused layouts (qkv_proj
weight+bias, gate_up) and carves each rank's TP shard while reading (head-sharded qkv,
row-sharded o/down, per-half-sharded gate_up — the layout
parallel/tp.py documents)."""
from __future__ import annotations

import glob
import os
from typing import Dict, Optional

import torch

from ..parallel import dist as pdist
from .config import ModelSpec


def _shard(t: torch.Tensor, dim: int, rank: int, tp: int) -> torch.Tensor:
    if tp == 1:
        return t
    n = t.shape[dim]
    assert n % tp == 0, (t.shape, dim, tp)
    return t.narrow(dim, rank * (n // tp), n // tp)


@torch.inference_mode()
def load_llama_safetensors(model, path: str, strict: bool = True) -> int:
    """Load a HF-layout checkpoint directory into a LlamaForCausalLM.
    Returns the number of tensors consumed."""
    from safetensors import safe_open

    spec: ModelSpec = model.spec
    tp = pdist.tp_world_size()
    rank = pdist.tp_rank()
    D = spec.head_dim
    dtype = model.embed.weight.dtype
    device = model.embed.weight.device

    files = sorted(glob.glob(os.path.join(path, "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no *.safetensors under {path}")
    tensors: Dict[str, torch.Tensor] = {}
    consumed = 0

    def want(name: str) -> bool:
        return name.startswith(("model.", "lm_head."))

    handles = []
    for f in files:
        h = safe_open(f, framework="pt", device="cpu")
        handles.append(h)
        for name in h.keys():
            if want(name):
                tensors[name] = h  # lazy: remember which file holds it

    def get(name: str) -> Optional[torch.Tensor]:
        h = tensors.get(name)
        return h.get_tensor(name) if h is not None else None

    def put(param: torch.Tensor, t: torch.Tensor) -> None:
        nonlocal consumed
        assert param.shape == t.shape, (param.shape, t.shape)
        param.copy_(t.to(dtype))
        consumed += 1

    emb = get("model.embed_tokens.weight")
    if emb is not None:
        put(model.embed.weight, emb)
    fin = get("model.norm.weight")
    if fin is not None:
        put(model.final_norm, fin)
    lm = get("lm_head.weight")
    if lm is not None and model.lm_head is not None:
        put(model.lm_head.weight, lm)

    for i, layer in enumerate(model.layers):
        pre = f"model.layers.{i}."
        ln1 = get(pre + "input_layernorm.weight")
        ln2 = get(pre + "post_attention_layernorm.weight")
        if ln1 is None and strict:
            raise KeyError(pre + "input_layernorm.weight")
        if ln1 is not None:
            put(layer.input_norm, ln1)
        if ln2 is not None:
            put(layer.post_attn_norm, ln2)
        q = get(pre + "self_attn.q_proj.weight")
        k = get(pre + "self_attn.k_proj.weight")
        v = get(pre + "self_attn.v_proj.weight")
        o = get(pre + "self_attn.o_proj.weight")
        if q is not None:
            fused = torch.cat([
                _shard(q, 0, rank, tp), _shard(k, 0, rank, tp),
                _shard(v, 0, rank, tp)])
            put(layer.attn.qkv_proj.weight, fused)
            put(layer.attn.o_proj.weight, _shard(o, 1, rank, tp))
        if layer.attn.qkv_proj.bias is not None:
            # Qwen2-family: per-projection biases fuse/shard like the rows
            qb = get(pre + "self_attn.q_proj.bias")
            kb = get(pre + "self_attn.k_proj.bias")
            vb = get(pre + "self_attn.v_proj.bias")
            if qb is None and strict:
                raise KeyError(pre + "self_attn.q_proj.bias")
            if qb is not None:
                put(layer.attn.qkv_proj.bias, torch.cat([
                    _shard(qb, 0, rank, tp), _shard(kb, 0, rank, tp),
                    _shard(vb, 0, rank, tp)]))
        if spec.n_experts == 0:
            g = get(pre + "mlp.gate_proj.weight")
            u = get(pre + "mlp.up_proj.weight")
            d = get(pre + "mlp.down_proj.weight")
            if g is not None:
                put(layer.mlp.gate_up.weight, torch.cat(
                    [_shard(g, 0, rank, tp), _shard(u, 0, rank, tp)]))
                put(layer.mlp.down.weight, _shard(d, 1, rank, tp))
        else:
            # Mixtral: block_sparse_moe.gate + experts.N.w1/w3/w2
            gate = get(pre + "block_sparse_moe.gate.weight")
            if gate is not None:
                put(layer.mlp.router.weight, gate)
            for e in range(spec.n_experts):
                epre = pre + f"block_sparse_moe.experts.{e}."
                w1 = get(epre + "w1.weight")  # gate
                w3 = get(epre + "w3.weight")  # up
                w2 = get(epre + "w2.weight")  # down
                if w1 is not None:
                    put(layer.mlp.experts[e].gate_up.weight, torch.cat(
                        [_shard(w1, 0, rank, tp), _shard(w3, 0, rank, tp)]))
                    put(layer.mlp.experts[e].down.weight, _shard(w2, 1, rank, tp))

    if model.lm_head is None and emb is not None:
        pass  # tied embeddings
    return consumed


def save_llama_safetensors(model, path: str) -> str:
    """Inverse mapping (TP=1 only): write this model's weights in HF layout
    (round-trip tested; also lets users export random-init fixtures)."""
    from safetensors.torch import save_file

    assert pdist.tp_world_size() == 1, "export from a TP=1 model"
    spec = model.spec
    D = spec.head_dim
    out: Dict[str, torch.Tensor] = {
        "model.embed_tokens.weight": model.embed.weight.float().cpu(),
        "model.norm.weight": model.final_norm.float().cpu(),
    }
    if model.lm_head is not None:
        out["lm_head.weight"] = model.lm_head.weight.float().cpu()
    for i, layer in enumerate(model.layers):
        pre = f"model.layers.{i}."
        out[pre + "input_layernorm.weight"] = layer.input_norm.float().cpu()
        out[pre + "post_attention_layernorm.weight"] = layer.post_attn_norm.float().cpu()
        qkv = layer.attn.qkv_proj.weight.float().cpu()
        nq = spec.n_heads * D
        nkv = spec.n_kv_heads * D
        out[pre + "self_attn.q_proj.weight"] = qkv[:nq]
        out[pre + "self_attn.k_proj.weight"] = qkv[nq:nq + nkv]
        out[pre + "self_attn.v_proj.weight"] = qkv[nq + nkv:]
        out[pre + "self_attn.o_proj.weight"] = layer.attn.o_proj.weight.float().cpu()
        if layer.attn.qkv_proj.bias is not None:  # Qwen2-family
            qkvb = layer.attn.qkv_proj.bias.float().cpu()
            out[pre + "self_attn.q_proj.bias"] = qkvb[:nq]
            out[pre + "self_attn.k_proj.bias"] = qkvb[nq:nq + nkv]
            out[pre + "self_attn.v_proj.bias"] = qkvb[nq + nkv:]
        if spec.n_experts == 0:
            gu = layer.mlp.gate_up.weight.float().cpu()
            I = spec.intermediate_size
            out[pre + "mlp.gate_proj.weight"] = gu[:I]
            out[pre + "mlp.up_proj.weight"] = gu[I:]
            out[pre + "mlp.down_proj.weight"] = layer.mlp.down.weight.float().cpu()
        else:
            out[pre + "block_sparse_moe.gate.weight"] = \
                layer.mlp.router.weight.float().cpu()
            for e, ex in enumerate(layer.mlp.experts):
                epre = pre + f"block_sparse_moe.experts.{e}."
                gu = ex.gate_up.weight.float().cpu()
                I = spec.intermediate_size
                out[epre + "w1.weight"] = gu[:I]
                out[epre + "w3.weight"] = gu[I:]
                out[epre + "w2.weight"] = ex.down.weight.float().cpu()
    os.makedirs(path, exist_ok=True)
    fp = os.path.join(path, "model.safetensors")
    save_file(out, fp)
    return fp
