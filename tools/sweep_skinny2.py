"""Cold-weight skinny-GEMM sweep: rotates >LLC-sized W copies so every
iteration streams W from HBM (the in-engine regime; the earlier sweep kept
one W hot in LLC and mistuned s_target).  Prints TB/s per config vs
hipBLASLt on the llama3-8b decode shapes."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from aurora_amd import _hip_ops as hip

import json
_env = os.environ.get("SHAPES")
SHAPES = json.loads(_env) if _env else [  # (tag, M, K, N)
    ("qkv", 64, 4096, 6144),
    ("o", 64, 4096, 4096),
    ("gate_up", 64, 4096, 28672),
    ("down", 64, 14336, 4096),
    ("qkv16", 16, 4096, 6144),
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

for tag, M, K, N in SHAPES:
    wb = N * K * 2
    ncopy = max(2, (LLC + wb - 1) // wb)
    ws = [torch.randn(N, K, device="cuda", dtype=torch.bfloat16) for _ in range(ncopy)]
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    iters = max(20, min(60, 3 * ncopy))
    res = {}
    only_blaslt = os.environ.get("BLASLT_ONLY") == "1"
    for st in () if (only_blaslt or M > 64) else (256, 512, 1024, 2048):
        for unr in (2, 4):
            dt = timeit(lambda i: hip.skinny_gemm_ex(x, ws[i % ncopy], st, unr), iters)
            res[f"sk_s{st}_u{unr}"] = dt
    if K % 128 == 0 and not only_blaslt:
        for st in (256, 512, 768, 1024, 2048):
            dt = timeit(lambda i: hip.skinny_gemm2_ex(x, ws[i % ncopy], st), iters)
            res[f"v2_s{st}"] = dt
        # numerics: v2 vs fp32 reference on the last W
        ref = (x.float() @ ws[0].float().t())
        got = hip.skinny_gemm2_ex(x, ws[0], 768).float()
        err = (got - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
        assert err < 2e-2, f"v2 numerics off: rel {err}"
    res["blaslt"] = timeit(lambda i: torch.matmul(x, ws[i % ncopy].t()), iters)
    best = min(res, key=res.get)
    print(f"== {tag} M={M} K={K} N={N} W={wb/1e6:.0f}MB (x{ncopy} copies)")
    for k in sorted(res, key=res.get):
        print(f"  {k:14s} {res[k]*1e6:8.1f} us  {wb/res[k]/1e12:5.2f} TB/s"
              + ("  <-- best" if k == best else ""))
    del ws
    torch.cuda.empty_cache()
