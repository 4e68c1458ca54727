"""aurora_amd — MI355X-native incident-investigation agent framework.

A from-scratch rebuild of the capabilities of Arvo-AI/aurora
designed MI355X-first: the agent's LLM runs in-process on a PyTorch-ROCm
engine whose hot ops (RMSNorm, RoPE, prefill/paged-decode attention, fused
SiLU-MLP, sampling, kNN) are hand-written HIP/CDNA4 (gfx950) kernels; RAG is
an on-GPU embedding encoder + HBM-resident kNN; guardrails are host-side
Sigma regex plus a GPU classifier; scaling is RCCL over xGMI (TP for the
model, DP over concurrent investigations).

Layer map (mirrors SURVEY.md §1, rebuilt):
  ops/        HIP kernels + CPU fp32 reference (numerics oracle)
  models/     Llama-3 family (GQA), Mixtral MoE, BGE-style encoder
  engine/     paged KV cache, continuous-batching scheduler, LLM engine,
              tokenizer, constrained tool-JSON decoding
  parallel/   torch.distributed (RCCL/gloo) process groups, TP linear layers
  rag/        document store, GPU kNN, hybrid lexical+vector search
  guardrails/ Sigma signature engine, input rail (fail-closed), redaction
  agent/      ReAct loop, tool registry, skills, orchestrator, LLM manager
  service/    HTTP API (webhooks/incidents/SSE/metrics), WS chat, SQLite
              store with org scoping, background investigation worker
  graph/      infrastructure dependency graph (Memgraph-compatible surface)
"""

__version__ = "0.1.0"
