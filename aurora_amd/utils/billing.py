"""Usage pricing: cost attribution for self-hosted inference.

The reference prices rented tokens per provider (reference:
utils/llm_usage_tracker.py, openrouter_pricing_service.py,
provider_pricing_service.py, billing_cache.py).  A self-hosted MI355X has
no per-token list price; cost = GPU-seconds x an hourly rate, attributed
to orgs by their share of generated work.  MODEL_RATES maps model ->
(prompt, completion) $/1M-token equivalents derived from measured
throughput at the configured GPU $/hr so dashboards stay comparable."""
from __future__ import annotations

import os
from typing import Dict

GPU_HOURLY_USD = float(os.environ.get("AURORA_GPU_HOURLY_USD", "2.50"))

# measured round-1 throughput on one MI355X (tokens/s whole-engine):
# llama3-8b decode ~ (64 seqs / ~8 ms/step) and prefill ~30k tok/s
_THROUGHPUT = {
    "llama3-8b": {"prompt": 30000.0, "completion": 8000.0},
    "llama3-70b": {"prompt": 4000.0, "completion": 1200.0},
    "mixtral-8x7b": {"prompt": 15000.0, "completion": 4000.0},
}
_DEFAULT = {"prompt": 20000.0, "completion": 5000.0}


def rate_per_mtok(model: str) -> Dict[str, float]:
    tp = _THROUGHPUT.get(model, _DEFAULT)
    per_sec = GPU_HOURLY_USD / 3600.0
    return {k: per_sec / v * 1e6 for k, v in tp.items()}


def cost_usd(model: str, prompt_tokens: int, completion_tokens: int) -> float:
    r = rate_per_mtok(model)
    return (prompt_tokens * r["prompt"] + completion_tokens * r["completion"]) / 1e6


def usage_with_cost(usage_summary: Dict[str, dict]) -> Dict[str, dict]:
    out = {}
    for model, row in usage_summary.items():
        out[model] = dict(row, cost_usd=round(
            cost_usd(model, row.get("pt") or 0, row.get("ct") or 0), 6))
    return out
