import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from aurora_amd.engine.engine import LLMEngine, EngineConfig
from aurora_amd.engine.sampling_params import SamplingParams
B = int(os.environ.get("B", "16"))
CTX = int(os.environ.get("CTX", "512"))
eng = LLMEngine(EngineConfig(model="llama3-8b", seed=0, kv_memory_fraction=0.45,
                             max_num_seqs=B + 2))
p = SamplingParams(temperature=0.0, max_tokens=400, ignore_eos=True)
for i in range(B):
    eng.add_request(f"r{i}", [30 + (j % 500) for j in range(CTX)], p)
for _ in range(40 + CTX // 256):
    eng.step()
torch.cuda.synchronize()
import time
t0 = time.perf_counter()
N = 60
for _ in range(N):
    eng.step()
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / N * 1000
print(f"done {eng.counters} wall_ms_per_step={dt:.2f}")
print("timers", {k: round(v, 2) for k, v in eng.timers.items()})
