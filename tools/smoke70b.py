"""Single-GPU Llama-3-70B feasibility: 140 GB bf16 weights in 288 GB HBM."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from aurora_amd.engine.engine import LLMEngine, EngineConfig
from aurora_amd.engine.sampling_params import SamplingParams

t0 = time.time()
eng = LLMEngine(EngineConfig(model="llama3-70b", kv_memory_fraction=0.15, seed=0))
print(f"init {time.time()-t0:.1f}s; free after init:",
      torch.cuda.mem_get_info()[0] / 2**30, "GiB")
t0 = time.time()
out = eng.generate([list(range(50, 306))],
                   SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True))
dt = time.time() - t0
print("decoded:", out[0], f"in {dt:.2f}s")
# quick decode rate: 16 more tokens timed
t0 = time.time()
out = eng.generate([list(range(50, 306))],
                   SamplingParams(temperature=0.0, max_tokens=16, ignore_eos=True))
print(f"llama3-70b 1-GPU: {16/(time.time()-t0):.1f} tok/s at batch 1")
