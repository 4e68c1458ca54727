"""Micro-bench the decode-step elementwise kernels at the headline batch
(B=256) against torch-native ceilings — rocprof showed them far off the
stream limit in-engine (profiles/decode_stats_b256 notes)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from aurora_amd import _hip_ops as hip

T = int(os.environ.get("T", "256"))
H, I, Hq, Hkv, D, P = 4096, 14336, 32, 8, 128, 256

def timeit(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

x = torch.randn(T, H, device="cuda", dtype=torch.bfloat16)
res = torch.randn(T, H, device="cuda", dtype=torch.bfloat16)
wt = torch.randn(H, device="cuda", dtype=torch.bfloat16)
gu = torch.randn(T, 2 * I, device="cuda", dtype=torch.bfloat16)
qkv = torch.randn(T, (Hq + 2 * Hkv) * D, device="cuda", dtype=torch.bfloat16)
pos = torch.arange(T, device="cuda", dtype=torch.int32)
slots = torch.arange(T, device="cuda", dtype=torch.int32)
cs = torch.randn(8192, D, device="cuda", dtype=torch.float32)
NP = (T + P - 1) // P + 1
kc = torch.zeros(NP, Hkv, P, D, device="cuda", dtype=torch.bfloat16)
vc = torch.zeros_like(kc)

MB = 1e6
r = {}
r["rmsnorm_res"] = (timeit(lambda: hip.rmsnorm_residual(x, res, wt, 1e-5)),
                    T * H * 2 * 4 / MB)
r["rmsnorm"] = (timeit(lambda: hip.rmsnorm(x, wt, 1e-5)), T * H * 2 * 2 / MB)


def torch_rms():
    h = (x + res).float()
    o = h * torch.rsqrt(h.pow(2).mean(-1, keepdim=True) + 1e-5)
    return (o * wt.float()).to(torch.bfloat16)

r["rmsnorm_torch"] = (timeit(torch_rms), T * H * 2 * 4 / MB)
r["silu_mul"] = (timeit(lambda: hip.silu_mul(gu)), T * I * 2 * 3 / MB)


def torch_silu():
    g, u = gu.chunk(2, -1)
    return torch.nn.functional.silu(g) * u

r["silu_torch"] = (timeit(torch_silu), T * I * 2 * 3 / MB)
r["rope_kv"] = (timeit(lambda: hip.fused_rope_kv_write(qkv, pos, slots, cs,
                                                       kc, vc, Hq, Hkv)),
                T * (Hq + 2 * Hkv) * D * 2 * 2 / MB)
r["memcpy_ref"] = (timeit(lambda: x.clone()), T * H * 2 * 2 / MB)
for k, (us, mb) in r.items():
    print(f"{k:14s} {us:8.1f} us  {mb/us*1e6/1e12:6.2f} TB/s  ({mb:.1f} MB)")
