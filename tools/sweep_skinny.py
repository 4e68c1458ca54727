import os, sys, time, json
sys.path.insert(0, "/root/repo")
import torch
from aurora_amd import _hip_ops

def timeit(fn, iters=30, warmup=8):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

dev = "cuda"
shapes = [("qkv", 6144, 4096), ("o", 4096, 4096), ("gateup", 28672, 4096),
          ("down", 4096, 14336), ("lmhead", 128256, 4096)]
res = {}
for tag, N, K in shapes:
    a = torch.randn(16, K, device=dev, dtype=torch.bfloat16)
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    gb = N * K * 2 / 1e9
    row = {"blaslt": round(gb / timeit(lambda: a @ w.t()), 2)}
    for st in (256, 512, 1024, 2048, 4096):
        for un in (2, 4):
            try:
                t = timeit(lambda: _hip_ops.skinny_gemm_ex(a, w, st, un))
                row[f"s{st}_u{un}"] = round(gb / t, 2)
            except Exception as e:
                row[f"s{st}_u{un}"] = str(e)[:40]
    res[tag] = row
    print(tag, json.dumps(row), flush=True)
