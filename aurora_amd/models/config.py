"""Model specs for the families the framework serves.

The reference outsources all model capability to provider APIs
(reference: server/chat/backend/agent/llm.py:32-67 — a registry of rented
models); here the registry is of locally-served architectures.  Weights are
random-initialised in this environment (no network); real deployments load
safetensors checkpoints via models.loader.
"""
from __future__ import annotations

from dataclasses import dataclass


@dataclass
class ModelSpec:
    name: str
    hidden_size: int
    n_layers: int
    n_heads: int
    n_kv_heads: int
    head_dim: int
    intermediate_size: int
    vocab_size: int
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_position: int = 131072
    tie_embeddings: bool = False
    # MoE (Mixtral-style); n_experts == 0 => dense MLP
    n_experts: int = 0
    experts_per_token: int = 2
    # Qwen2-family attention projections carry a bias (q/k/v only)
    qkv_bias: bool = False
    # encoder-only (BGE-style) models use bidirectional attention + pooling
    is_encoder: bool = False

    @property
    def q_size(self) -> int:
        return self.n_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.n_kv_heads * self.head_dim

    def param_count(self) -> int:
        h, i, v, L = self.hidden_size, self.intermediate_size, self.vocab_size, self.n_layers
        attn = h * (self.q_size + 2 * self.kv_size) + self.q_size * h
        mlp = 3 * h * i if self.n_experts == 0 else self.n_experts * 3 * h * i
        emb = v * h * (1 if self.tie_embeddings else 2)
        return L * (attn + mlp) + emb


_REGISTRY: dict[str, ModelSpec] = {}


def register(spec: ModelSpec) -> ModelSpec:
    _REGISTRY[spec.name] = spec
    return spec


def get_spec(name: str) -> ModelSpec:
    try:
        return _REGISTRY[name]
    except KeyError:
        raise KeyError(f"unknown model spec '{name}'; known: {sorted(_REGISTRY)}") from None


# ---- presets (BASELINE.json configs) -------------------------------------

register(ModelSpec(  # BASELINE config 2/5 agent model
    name="llama3-8b", hidden_size=4096, n_layers=32, n_heads=32, n_kv_heads=8,
    head_dim=128, intermediate_size=14336, vocab_size=128256,
))
register(ModelSpec(
    name="llama3-70b", hidden_size=8192, n_layers=80, n_heads=64, n_kv_heads=8,
    head_dim=128, intermediate_size=28672, vocab_size=128256,
))
register(ModelSpec(
    name="llama3.2-1b", hidden_size=2048, n_layers=16, n_heads=32, n_kv_heads=8,
    head_dim=64, intermediate_size=8192, vocab_size=128256, tie_embeddings=True,
))
register(ModelSpec(  # BASELINE config 4 (MoE tool-calling agent)
    name="mixtral-8x7b", hidden_size=4096, n_layers=32, n_heads=32, n_kv_heads=8,
    head_dim=128, intermediate_size=14336, vocab_size=32000, rope_theta=1e6,
    n_experts=8, experts_per_token=2,
))
register(ModelSpec(  # Qwen2/2.5-7B: llama arch + QKV bias, 1e6 rope, 152k vocab
    name="qwen2-7b", hidden_size=3584, n_layers=28, n_heads=28, n_kv_heads=4,
    head_dim=128, intermediate_size=18944, vocab_size=152064, rope_theta=1e6,
    max_position=32768, rms_eps=1e-6, qkv_bias=True,
))
register(ModelSpec(
    name="qwen2-72b", hidden_size=8192, n_layers=80, n_heads=64, n_kv_heads=8,
    head_dim=128, intermediate_size=29568, vocab_size=152064, rope_theta=1e6,
    max_position=32768, rms_eps=1e-6, qkv_bias=True,
))
register(ModelSpec(  # BGE-base-style embedding encoder (RAG path)
    name="bge-base", hidden_size=768, n_layers=12, n_heads=12, n_kv_heads=12,
    head_dim=64, intermediate_size=3072, vocab_size=30522, is_encoder=True,
    max_position=512, rms_eps=1e-12,
))
# tiny configs for CPU tests / plumbing validation (BASELINE config 1)
register(ModelSpec(
    name="tiny-llama", hidden_size=64, n_layers=2, n_heads=4, n_kv_heads=2,
    head_dim=16, intermediate_size=128, vocab_size=512, max_position=2048,
    tie_embeddings=True,
))
register(ModelSpec(  # Qwen2-shaped tiny config (qkv_bias path on CPU)
    name="tiny-qwen", hidden_size=64, n_layers=2, n_heads=4, n_kv_heads=2,
    head_dim=16, intermediate_size=128, vocab_size=512, max_position=2048,
    rope_theta=1e6, qkv_bias=True, tie_embeddings=True,
))
register(ModelSpec(
    name="tiny-moe", hidden_size=64, n_layers=2, n_heads=4, n_kv_heads=2,
    head_dim=16, intermediate_size=96, vocab_size=512, max_position=2048,
    n_experts=4, experts_per_token=2, tie_embeddings=True,
))
register(ModelSpec(
    name="tiny-encoder", hidden_size=64, n_layers=2, n_heads=4, n_kv_heads=4,
    head_dim=16, intermediate_size=128, vocab_size=512, max_position=512,
    is_encoder=True,
))
register(ModelSpec(  # GPU smoke/test model: real head_dim for the HIP kernels
    name="smoke-llama", hidden_size=512, n_layers=4, n_heads=4, n_kv_heads=2,
    head_dim=128, intermediate_size=1024, vocab_size=1024, max_position=8192,
    tie_embeddings=True,
))
register(ModelSpec(  # GPU long-context smoke: 64k positions, tiny layers
    name="smoke-llama-64k", hidden_size=512, n_layers=4, n_heads=4,
    n_kv_heads=2, head_dim=128, intermediate_size=1024, vocab_size=1024,
    max_position=65536, tie_embeddings=True,
))
register(ModelSpec(  # GPU Qwen2-shaped smoke: qkv_bias on the HIP path
    name="smoke-qwen", hidden_size=512, n_layers=4, n_heads=4, n_kv_heads=2,
    head_dim=128, intermediate_size=1024, vocab_size=1024, max_position=8192,
    rope_theta=1e6, qkv_bias=True, tie_embeddings=True,
))
register(ModelSpec(  # GPU MoE smoke: real head_dim for the HIP kernels
    name="smoke-moe", hidden_size=512, n_layers=4, n_heads=4, n_kv_heads=2,
    head_dim=128, intermediate_size=768, vocab_size=1024, max_position=8192,
    n_experts=4, experts_per_token=2, tie_embeddings=True,
))
