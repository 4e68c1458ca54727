import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from aurora_amd.engine.engine import LLMEngine, EngineConfig
from aurora_amd.engine.sampling_params import SamplingParams
eng = LLMEngine(EngineConfig(model="llama3-8b", seed=0, kv_memory_fraction=0.3))
p = SamplingParams(temperature=0.0, max_tokens=400, ignore_eos=True)
for i in range(16):
    eng.add_request(f"r{i}", list(range(30, 542)), p)
for _ in range(40):
    eng.step()
torch.cuda.synchronize()
for _ in range(100):
    eng.step()
torch.cuda.synchronize()
print("done", eng.counters)
