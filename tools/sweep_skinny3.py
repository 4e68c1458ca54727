"""Cold-weight sweep for the v3 (M=128-256) projection GEMM: numerics vs
fp32 torch reference, then TB/s vs hipBLASLt and v2m on the llama3-8b
decode shapes with W copies rotated past the 256 MB LLC."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from aurora_amd import _hip_ops as hip

_env = os.environ.get("SHAPES")
SHAPES = json.loads(_env) if _env else [  # (tag, M, K, N)
    ("qkv", 256, 4096, 6144),
    ("o", 256, 4096, 4096),
    ("gate_up", 256, 4096, 28672),
    ("down", 256, 14336, 4096),
    ("qkv128", 128, 4096, 6144),
    ("down128", 128, 14336, 4096),
]
LLC = 300 * 1024 * 1024


def timeit(fn, iters):
    for _ in range(3):
        fn(0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(iters):
        fn(i)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


# ---- numerics first (any failure must abort before perf) ----
for M, K, N in ((256, 4096, 6144), (256, 1024, 256), (192, 512, 320),
                (130, 4096, 4096), (256, 14336, 4096)):
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    ref = x.float() @ w.float().t()
    for st in (0, 1, 256):
        got = hip.skinny_gemm3_ex(x, w, st).float()
        err = (got - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
        assert err < 2e-2, f"v3 numerics M={M} K={K} N={N} st={st}: rel {err}"
print("v3 numerics OK")

for tag, M, K, N in SHAPES:
    wb = N * K * 2
    ncopy = max(2, (LLC + wb - 1) // wb)
    ws = [torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
          for _ in range(ncopy)]
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    iters = max(20, min(60, 3 * ncopy))
    res = {}
    for st in (256, 384, 512, 640, 896):
        dt = timeit(lambda i: hip.skinny_gemm3_ex(x, ws[i % ncopy], st), iters)
        res[f"v3_s{st}"] = dt
    dt = timeit(lambda i: hip.skinny_gemm2_ex(x, ws[i % ncopy], 512), iters)
    res["v2m_s512"] = dt
    res["blaslt"] = timeit(lambda i: torch.matmul(x, ws[i % ncopy].t()), iters)
    best = min(res, key=res.get)
    print(f"== {tag} M={M} K={K} N={N} W={wb/1e6:.0f}MB (x{ncopy} copies)")
    for k in sorted(res, key=res.get):
        print(f"  {k:12s} {res[k]*1e6:8.1f} us  {wb/res[k]/1e12:5.2f} TB/s"
              + ("  <-- best" if k == best else ""))
    del ws
    torch.cuda.empty_cache()
