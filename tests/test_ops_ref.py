"""Internal-consistency tests of the fp32 reference ops (CPU).

These pin the semantics the HIP kernels are tested against (tests/test_hip_kernels.py).
"""
import math

import pytest
import torch

from aurora_amd.ops import torch_ref as R


def test_rmsnorm_matches_manual():
    x = torch.randn(7, 32)
    w = torch.randn(32)
    y = R.rmsnorm(x, w, 1e-5)
    ref = x / (x.pow(2).mean(-1, keepdim=True) + 1e-5).sqrt() * w
    assert torch.allclose(y, ref, atol=1e-5)


def test_rmsnorm_residual():
    x, res = torch.randn(5, 16), torch.randn(5, 16)
    w = torch.randn(16)
    y, new_res = R.rmsnorm_residual(x, res, w)
    assert torch.allclose(new_res, x + res, atol=1e-6)
    assert torch.allclose(y, R.rmsnorm(x + res, w), atol=1e-6)


def test_rope_norm_preserving_and_position_zero():
    D = 16
    cs = R.rope_cos_sin_cache(D, 32, theta=10000.0)
    q = torch.randn(4, 2, D)
    k = torch.randn(4, 1, D)
    pos = torch.tensor([0, 1, 5, 31])
    q2, k2 = R.rope_apply(q, k, pos, cs)
    # position 0 is identity
    assert torch.allclose(q2[0], q[0], atol=1e-6)
    # rotation preserves per-pair norms
    assert torch.allclose(q2.norm(dim=-1), q.norm(dim=-1), atol=1e-5)
    # nonzero positions actually rotate
    assert not torch.allclose(q2[2], q[2])


def test_silu_mul():
    gu = torch.randn(6, 20)
    y = R.silu_mul(gu)
    g, u = gu[:, :10], gu[:, 10:]
    assert torch.allclose(y, torch.nn.functional.silu(g) * u, atol=1e-6)


def test_attention_prefill_varlen_vs_sdpa():
    torch.manual_seed(0)
    Hq, Hkv, D = 4, 2, 16
    lens = [5, 9]
    T = sum(lens)
    q = torch.randn(T, Hq, D)
    k = torch.randn(T, Hkv, D)
    v = torch.randn(T, Hkv, D)
    cu = torch.tensor([0, 5, 14], dtype=torch.int32)
    out = R.attention_prefill(q, k, v, cu, 1.0 / math.sqrt(D), causal=True)
    # per-seq dense SDPA reference
    a = 0
    for L in lens:
        qs = q[a:a+L].transpose(0, 1)
        ks = k[a:a+L].repeat_interleave(2, 1).transpose(0, 1)
        vs = v[a:a+L].repeat_interleave(2, 1).transpose(0, 1)
        ref = torch.nn.functional.scaled_dot_product_attention(qs, ks, vs, is_causal=True)
        assert torch.allclose(out[a:a+L], ref.transpose(0, 1), atol=1e-5)
        a += L


def test_paged_prefill_and_decode_consistent_with_dense():
    """Write K/V through the cache and check paged prefill+decode against the
    self-contained varlen attention."""
    torch.manual_seed(0)
    Hq, Hkv, D, P = 4, 2, 16, 8
    L = 21
    q = torch.randn(L, Hq, D)
    k = torch.randn(L, Hkv, D)
    v = torch.randn(L, Hkv, D)
    kc = torch.zeros(8, Hkv, P, D)
    vc = torch.zeros(8, Hkv, P, D)
    pages = [3, 0, 5]
    slots = torch.tensor([pages[i // P] * P + i % P for i in range(L)], dtype=torch.int32)
    R.kv_cache_write(k, v, kc, vc, slots)
    bt = torch.tensor([pages], dtype=torch.int32)
    cu = torch.tensor([0, L], dtype=torch.int32)
    ctx = torch.tensor([L], dtype=torch.int32)
    scale = 1.0 / math.sqrt(D)
    out_paged = R.attention_prefill_paged(q, kc, vc, bt, cu, ctx, scale)
    out_dense = R.attention_prefill(q, k, v, cu, scale, causal=True)
    assert torch.allclose(out_paged, out_dense, atol=1e-5)
    # decode of the last token == last row of prefill
    out_dec = R.paged_attention_decode(q[-1:], kc, vc, bt, ctx, scale)
    assert torch.allclose(out_dec[0], out_dense[-1], atol=1e-5)


def test_append_prefill_matches_full():
    """Append-prefill of a chunk against cached history == rows of full prefill."""
    torch.manual_seed(1)
    Hq, Hkv, D, P = 2, 1, 8, 4
    L, chunk = 10, 3
    q = torch.randn(L, Hq, D)
    k = torch.randn(L, Hkv, D)
    v = torch.randn(L, Hkv, D)
    kc = torch.zeros(4, Hkv, P, D)
    vc = torch.zeros(4, Hkv, P, D)
    pages = [1, 2, 0]
    slots = torch.tensor([pages[i // P] * P + i % P for i in range(L)], dtype=torch.int32)
    R.kv_cache_write(k, v, kc, vc, slots)
    bt = torch.tensor([pages], dtype=torch.int32)
    scale = 1.0 / math.sqrt(D)
    full = R.attention_prefill(q, k, v, torch.tensor([0, L], dtype=torch.int32), scale)
    app = R.attention_prefill_paged(
        q[L - chunk:], kc, vc, bt,
        torch.tensor([0, chunk], dtype=torch.int32),
        torch.tensor([L], dtype=torch.int32), scale,
    )
    assert torch.allclose(app, full[L - chunk:], atol=1e-5)


def test_embed_pool_normalized():
    h = torch.randn(12, 8)
    cu = torch.tensor([0, 4, 12], dtype=torch.int32)
    e = R.embed_pool(h, cu)
    assert e.shape == (2, 8)
    assert torch.allclose(e.norm(dim=-1), torch.ones(2), atol=1e-5)


def test_knn_topk():
    docs = torch.nn.functional.normalize(torch.randn(100, 16), dim=-1)
    qv = docs[17:18] + 0.01 * torch.randn(1, 16)
    vals, idx = R.knn_topk(torch.nn.functional.normalize(qv, dim=-1), docs, 5)
    assert idx[0, 0] == 17


def test_softmax_sample_greedy_and_topk():
    logits = torch.tensor([[0.0, 5.0, 1.0], [3.0, 0.0, 0.0]])
    out = R.softmax_sample(logits, torch.zeros(2), torch.ones(2), torch.zeros(2, dtype=torch.int64))
    assert out.tolist() == [1, 0]
    g = torch.Generator().manual_seed(0)
    out = R.softmax_sample(
        logits, torch.full((2,), 1.0), torch.ones(2), torch.ones(2, dtype=torch.int64), g
    )
    assert out.tolist() == [1, 0]  # top_k=1 => argmax
