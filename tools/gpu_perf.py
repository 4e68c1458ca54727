#!/usr/bin/env python3
"""Kernel micro-benchmarks on a real MI355X (run under gpurun).

Reports achieved bandwidth/compute against the chip ceilings
(HBM ~6.3 TB/s achievable, bf16 MFMA ~2.38 PF µbench) so each kernel's
headroom is visible without a full rocprof pass.
"""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import json
import math
import time

import torch

from aurora_amd import ops


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    out = {}
    dev = "cuda"
    # ---- rmsnorm: llama3-8b shape, prefill-sized
    T, H = 8192, 4096
    x = torch.randn(T, H, device=dev, dtype=torch.bfloat16)
    w = torch.randn(H, device=dev, dtype=torch.bfloat16)
    t = timeit(lambda: ops.rmsnorm(x, w, 1e-5))
    bytes_moved = T * H * 2 * 2  # read + write
    out["rmsnorm_8192x4096"] = {"us": t * 1e6, "GBps": bytes_moved / t / 1e9}

    r = torch.randn_like(x)
    t = timeit(lambda: ops.rmsnorm_residual(x, r, w, 1e-5))
    out["rmsnorm_residual"] = {"us": t * 1e6, "GBps": T * H * 2 * 4 / t / 1e9}

    # ---- silu_mul: [8192, 2*14336]
    gu = torch.randn(8192, 2 * 14336, device=dev, dtype=torch.bfloat16)
    t = timeit(lambda: ops.silu_mul(gu))
    out["silu_mul_8192x14336"] = {"us": t * 1e6,
                                  "GBps": 8192 * 14336 * 3 * 2 / t / 1e9}

    # ---- paged decode attention: llama3-8b, B=16, ctx=4096
    B, Hq, Hkv, D, P = 16, 32, 8, 128, 256
    ctxv = 4096
    NP = B * ctxv // P + B
    kc = torch.randn(NP, Hkv, P, D, device=dev, dtype=torch.bfloat16)
    vc = torch.randn_like(kc)
    q = torch.randn(B, Hq, D, device=dev, dtype=torch.bfloat16)
    bt = torch.arange(NP, dtype=torch.int32, device=dev)[: B * (ctxv // P)].reshape(B, -1)
    ctx = torch.full((B,), ctxv, dtype=torch.int32, device=dev)
    scale = 1 / math.sqrt(D)
    t = timeit(lambda: ops.paged_attention_decode(q, kc, vc, bt, ctx, scale))
    kv_bytes = B * Hkv * ctxv * D * 2 * 2
    out[f"paged_decode_B{B}_ctx{ctxv}"] = {"us": t * 1e6, "KV_GBps": kv_bytes / t / 1e9}

    B3 = 2
    q3 = torch.randn(B3, Hq, D, device=dev, dtype=torch.bfloat16)
    ctx3 = torch.full((B3,), ctxv, dtype=torch.int32, device=dev)
    t = timeit(lambda: ops.paged_attention_decode(q3, kc, vc, bt[:B3], ctx3, scale))
    out[f"paged_decode_B{B3}_ctx{ctxv}"] = {
        "us": t * 1e6, "KV_GBps": B3 * Hkv * ctxv * D * 2 * 2 / t / 1e9}

    B2 = 64
    bt2 = bt.repeat(4, 1)
    q2 = torch.randn(B2, Hq, D, device=dev, dtype=torch.bfloat16)
    ctx2 = torch.full((B2,), ctxv, dtype=torch.int32, device=dev)
    t = timeit(lambda: ops.paged_attention_decode(q2, kc, vc, bt2, ctx2, scale))
    out[f"paged_decode_B{B2}_ctx{ctxv}"] = {
        "us": t * 1e6, "KV_GBps": B2 * Hkv * ctxv * D * 2 * 2 / t / 1e9}

    # ---- prefill attention: 1 seq of 4096, llama3-8b heads
    L = 4096
    qp = torch.randn(L, Hq, D, device=dev, dtype=torch.bfloat16)
    kn = torch.randn(L, Hkv, D, device=dev, dtype=torch.bfloat16)
    vn = torch.randn(L, Hkv, D, device=dev, dtype=torch.bfloat16)
    npg = L // P
    kcp = torch.zeros(npg + 1, Hkv, P, D, device=dev, dtype=torch.bfloat16)
    vcp = torch.zeros_like(kcp)
    slots = torch.arange(L, dtype=torch.int32, device=dev)
    ops.kv_cache_write(kn, vn, kcp, vcp, slots)
    btp = torch.arange(npg + 1, dtype=torch.int32, device=dev).unsqueeze(0)
    cu = torch.tensor([0, L], dtype=torch.int32, device=dev)
    ctxp = torch.tensor([L], dtype=torch.int32, device=dev)
    t = timeit(lambda: ops.attention_prefill_paged(qp, kcp, vcp, btp, cu, ctxp, scale))
    flops = 2 * 2 * Hq * D * L * L / 2  # QK^T + PV, causal half
    out[f"prefill_L{L}"] = {"us": t * 1e6, "TFLOPs": flops / t / 1e12}

    # batch-of-8 x 2048 (the bench-like shape)
    lens = [2048] * 8
    Lt = sum(lens)
    qp = torch.randn(Lt, Hq, D, device=dev, dtype=torch.bfloat16)
    kn = torch.randn(Lt, Hkv, D, device=dev, dtype=torch.bfloat16)
    vn = torch.randn(Lt, Hkv, D, device=dev, dtype=torch.bfloat16)
    npg_per = 2048 // P
    kcp = torch.zeros(8 * npg_per + 1, Hkv, P, D, device=dev, dtype=torch.bfloat16)
    vcp = torch.zeros_like(kcp)
    slots = torch.arange(Lt, dtype=torch.int32, device=dev)
    ops.kv_cache_write(kn, vn, kcp, vcp, slots)
    btp = torch.arange(8 * npg_per, dtype=torch.int32, device=dev).reshape(8, npg_per)
    cu = torch.arange(0, Lt + 1, 2048, dtype=torch.int32, device=dev)
    ctxp = torch.full((8,), 2048, dtype=torch.int32, device=dev)
    t = timeit(lambda: ops.attention_prefill_paged(qp, kcp, vcp, btp, cu, ctxp, scale))
    flops = 8 * 2 * 2 * Hq * D * 2048 * 2048 / 2
    out["prefill_8x2048"] = {"us": t * 1e6, "TFLOPs": flops / t / 1e12}

    # ---- GEMM reference point (hipBLASLt): llama3-8b qkv
    a = torch.randn(8192, 4096, device=dev, dtype=torch.bfloat16)
    b = torch.randn(6144, 4096, device=dev, dtype=torch.bfloat16)
    t = timeit(lambda: a @ b.t())
    out["gemm_8192x6144x4096"] = {"us": t * 1e6,
                                  "TFLOPs": 2 * 8192 * 6144 * 4096 / t / 1e12}

    # decode-shaped GEMV batch: [16, 4096] x [4096, 6144]
    a = torch.randn(16, 4096, device=dev, dtype=torch.bfloat16)
    t = timeit(lambda: a @ b.t())
    out["gemv_16x6144x4096_blaslt"] = {"us": t * 1e6,
                                       "GBps": 6144 * 4096 * 2 / t / 1e9}
    from aurora_amd import _hip_ops
    t = timeit(lambda: _hip_ops.skinny_gemm(a, b))
    out["gemv_16x6144x4096_skinny"] = {"us": t * 1e6,
                                       "GBps": 6144 * 4096 * 2 / t / 1e9}
    bd = torch.randn(4096, 14336, device=dev, dtype=torch.bfloat16)
    t = timeit(lambda: _hip_ops.skinny_gemm(a[:, :4096], bd.t().contiguous()))
    t2 = timeit(lambda: _hip_ops.skinny_gemm(torch.randn(16, 14336, device=dev, dtype=torch.bfloat16), bd))
    out["gemv_16x4096x14336_skinny_down"] = {"us": t2 * 1e6,
                                             "GBps": 14336 * 4096 * 2 / t2 / 1e9}
    for (nn, kk, tag) in [(4096, 4096, "o"), (28672, 4096, "gateup"),
                          (4096, 14336, "down")]:
        ww = torch.randn(nn, kk, device=dev, dtype=torch.bfloat16)
        aa = torch.randn(16, kk, device=dev, dtype=torch.bfloat16)
        tb = timeit(lambda: aa @ ww.t())
        ts = timeit(lambda: _hip_ops.skinny_gemm(aa, ww))
        out[f"gemv_16_{tag}"] = {"blaslt_GBps": nn * kk * 2 / tb / 1e9,
                                 "skinny_GBps": nn * kk * 2 / ts / 1e9}
    blm = torch.randn(128256, 4096, device=dev, dtype=torch.bfloat16)
    t = timeit(lambda: _hip_ops.skinny_gemm(a, blm), iters=10)
    out["gemv_16x128256x4096_skinny_lmhead"] = {"us": t * 1e6,
                                                "GBps": 128256 * 4096 * 2 / t / 1e9}
    t = timeit(lambda: a @ blm.t(), iters=10)
    out["gemv_16x128256x4096_blaslt_lmhead"] = {"us": t * 1e6,
                                                "GBps": 128256 * 4096 * 2 / t / 1e9}

    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
