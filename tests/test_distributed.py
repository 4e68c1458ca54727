"""Multi-process distributed tests (gloo backend, world_size=2, CPU).

The driver's 8-GPU scaling runs are out of our hands; these make the
distributed path correct by construction: TP=2 forward must equal the
single-process full model, and the DP bench contract must aggregate."""
import json
import os
import subprocess
import sys

import pytest
import torch
import torch.multiprocessing as mp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _tp_worker(rank: int, world: int, port: int, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    sys.path.insert(0, ROOT)
    import torch
    from aurora_amd.models.config import get_spec
    from aurora_amd.models.llama import AttnMetadata, LlamaForCausalLM
    from aurora_amd.parallel import dist as pdist

    # phase 1: TP=1 so the full (unsharded) model builds correctly
    pdist.init_distributed(tp_size=1, backend="gloo")
    spec = get_spec("tiny-llama")
    torch.manual_seed(0)
    full = LlamaForCausalLM(spec, dtype=torch.float32)
    full.init_random(7)
    D = spec.head_dim
    hq, hkv = spec.n_heads // world, spec.n_kv_heads // world
    inter = spec.intermediate_size // world

    def build_tp(full, spec, rank, world):
        from aurora_amd.models.llama import LlamaForCausalLM
        tp = LlamaForCausalLM(spec, dtype=torch.float32)
        _copy_shards(tp, full, spec, rank, world)
        return tp

    def _copy_shards(tp, full, spec, rank, world):
      with torch.no_grad():
            tp.embed.weight.copy_(full.embed.weight)
            tp.final_norm.copy_(full.final_norm)
            for lf, lt in zip(full.layers, tp.layers):
                lt.input_norm.copy_(lf.input_norm)
                lt.post_attn_norm.copy_(lf.post_attn_norm)
                wf = lf.attn.qkv_proj.weight  # [(Hq+2Hkv)*D, H] full
                q_w = wf[: spec.n_heads * D]
                k_w = wf[spec.n_heads * D: (spec.n_heads + spec.n_kv_heads) * D]
                v_w = wf[(spec.n_heads + spec.n_kv_heads) * D:]
                lt.attn.qkv_proj.weight.copy_(torch.cat([
                    q_w[rank * hq * D: (rank + 1) * hq * D],
                    k_w[rank * hkv * D: (rank + 1) * hkv * D],
                    v_w[rank * hkv * D: (rank + 1) * hkv * D]]))
                # o_proj: row-parallel (input sharded by head)
                lt.attn.o_proj.weight.copy_(
                    lf.attn.o_proj.weight[:, rank * hq * D: (rank + 1) * hq * D])
                gf = lf.mlp.gate_up.weight  # [2I, H]
                lt.mlp.gate_up.weight.copy_(torch.cat([
                    gf[rank * inter: (rank + 1) * inter],
                    gf[spec.intermediate_size + rank * inter:
                       spec.intermediate_size + (rank + 1) * inter]]))
                lt.mlp.down.weight.copy_(
                    lf.mlp.down.weight[:, rank * inter: (rank + 1) * inter])

    L = 12
    P = 16
    ids = torch.arange(5, 5 + L)

    def run(model, kvh):
        caches = [(torch.zeros(2, kvh, P, D), torch.zeros(2, kvh, P, D))
                  for _ in range(spec.n_layers)]
        meta = AttnMetadata(
            mode="prefill",
            positions=torch.arange(L, dtype=torch.int32),
            slot_mapping=torch.arange(L, dtype=torch.int32),
            block_tables=torch.arange(2, dtype=torch.int32).unsqueeze(0),
            context_lens=torch.tensor([L], dtype=torch.int32),
            cu_seqlens_q=torch.tensor([0, L], dtype=torch.int32),
        )
        h = model(ids, caches, meta)
        return model.compute_logits(h)

    full_logits = run(full, spec.n_kv_heads)
    # phase 2: switch to TP=2, build the sharded model, carve shards
    pdist.init_distributed(tp_size=world, backend="gloo")
    tp = build_tp(full, spec, rank, world)
    tp_logits = run(tp, hkv)
    err = (tp_logits - full_logits).abs().max().item()
    q.put((rank, err))


def _moe_tp_worker(rank: int, world: int, port: int, tmp: str, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    sys.path.insert(0, ROOT)
    import torch
    from aurora_amd.models.config import get_spec
    from aurora_amd.models.llama import AttnMetadata, LlamaForCausalLM
    from aurora_amd.models.loader import load_llama_safetensors, save_llama_safetensors
    from aurora_amd.parallel import dist as pdist

    # phase 1: full model on TP=1, exported through the real HF-layout writer
    pdist.init_distributed(tp_size=1, backend="gloo")
    spec = get_spec("smoke-moe")
    torch.manual_seed(1)
    full = LlamaForCausalLM(spec, dtype=torch.float32)
    full.init_random(11)
    ckpt = os.path.join(tmp, "moe_ckpt")
    if rank == 0:
        save_llama_safetensors(full, ckpt)
    torch.distributed.barrier()

    L, P, D = 9, 16, spec.head_dim

    def run(model, kvh):
        caches = [(torch.zeros(2, kvh, P, D), torch.zeros(2, kvh, P, D))
                  for _ in range(spec.n_layers)]
        meta = AttnMetadata(
            mode="prefill",
            positions=torch.arange(L, dtype=torch.int32),
            slot_mapping=torch.arange(L, dtype=torch.int32),
            block_tables=torch.arange(2, dtype=torch.int32).unsqueeze(0),
            context_lens=torch.tensor([L], dtype=torch.int32),
            cu_seqlens_q=torch.tensor([0, L], dtype=torch.int32),
        )
        h = model(torch.arange(3, 3 + L), caches, meta)
        return model.compute_logits(h)

    full_logits = run(full, spec.n_kv_heads)
    # phase 2: TP=2 model loaded through the loader's shard carving
    pdist.init_distributed(tp_size=world, backend="gloo")
    tp = LlamaForCausalLM(spec, dtype=torch.float32)
    load_llama_safetensors(tp, ckpt)
    tp_logits = run(tp, spec.n_kv_heads // world)
    q.put((rank, (tp_logits - full_logits).abs().max().item()))


@pytest.mark.timeout(300)
def test_tp2_moe_loader_matches_single_process(tmp_path):
    """Mixtral-style MoE under TP=2: expert weights carved by the real
    loader (models/loader.py _shard) + the batched-dispatch all-reduce
    must reproduce the unsharded forward exactly (fp32/gloo)."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29531
    ps = [ctx.Process(target=_moe_tp_worker, args=(r, 2, port, str(tmp_path), q))
          for r in range(2)]
    [p.start() for p in ps]
    errs = [q.get() for _ in range(2)]
    [p.join(timeout=240) for p in ps]
    for rank, err in errs:
        assert err < 1e-4, (rank, err)


@pytest.mark.timeout(300)
def test_tp2_matches_single_process():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29611
    procs = [ctx.Process(target=_tp_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, err in results:
        assert err < 1e-3, (rank, err)


@pytest.mark.timeout(300)
def test_bench_dp2_cpu():
    """The driver's multi-rank launch shape: torch.distributed.run, 2 CPU
    ranks, gloo; rank 0 prints one JSON line with whole-job aggregation."""
    env = dict(os.environ, BENCH_MODEL="tiny-llama", BENCH_CONCURRENCY="2",
               BENCH_TURNS="2", BENCH_RAG="0", MASTER_ADDR="127.0.0.1")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29612", "bench.py", "--gpus", "2", "--steps", "1",
         "--warmup", "0", "--device", "cpu", "--first-prompt", "32",
         "--turn-prompt", "16", "--decode-per-turn", "8"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=240)
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert lines, out.stdout + out.stderr
    doc = json.loads(lines[-1])
    assert doc["metric"] == "incidents/hr"
    assert doc["config"]["global_batch"] == 4  # 2 ranks x 2 incidents
    assert doc["config"]["parallelism"] == "dp2"
    assert doc["value"] > 0


@pytest.mark.timeout(300)
def test_bench_e2e_deadline_straggler_accounting():
    """run_step_e2e's deadline records stragglers (reported and excluded
    from the throughput numerator) instead of hanging the bench forever."""
    import asyncio
    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "bench_mod", os.path.join(ROOT, "bench.py"))
    bench = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(bench)

    import torch as _t
    from aurora_amd.engine.engine import EngineConfig
    from aurora_amd.engine.server import AsyncEngineServer
    from aurora_amd.service.runtime import Runtime
    from aurora_amd.utils.config import Settings

    class Args:
        concurrency = 2
        turns = 2
        rag_docs = 5
        model = "tiny-llama"

    async def main():
        cfg = Settings(db_path=":memory:",
                       secrets_path="/tmp/bench_dl_sec.json")
        cfg.worker_concurrency = 0  # workers never start -> nothing completes
        srv = AsyncEngineServer(EngineConfig(model="tiny-llama",
                                             device="cpu", seed=5))
        rt = Runtime(cfg, engine_server=srv)
        org = rt.admin.create_org("dl")
        rng = _t.Generator().manual_seed(1)
        lat, statuses = await bench.run_step_e2e(rt, org, Args(), rng, 0,
                                                 deadline_s=3.0)
        assert statuses.get("straggler", 0) == 2  # both stragglers accounted
        assert lat == []
        await rt.stop()
        srv.shutdown()

    asyncio.run(main())


def _qwen_tp_worker(rank: int, world: int, port: int, tmp: str, q):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    sys.path.insert(0, ROOT)
    import torch
    from aurora_amd.models.config import get_spec
    from aurora_amd.models.llama import AttnMetadata, LlamaForCausalLM
    from aurora_amd.models.loader import (load_llama_safetensors,
                                          save_llama_safetensors)
    from aurora_amd.parallel import dist as pdist

    # phase 1: full Qwen2-shaped model (qkv_bias) exported at TP=1
    pdist.init_distributed(tp_size=1, backend="gloo")
    spec = get_spec("tiny-qwen")
    torch.manual_seed(2)
    full = LlamaForCausalLM(spec, dtype=torch.float32)
    full.init_random(21)
    # non-constant bias so a sharding mistake cannot cancel out
    gen = torch.Generator().manual_seed(77)
    for layer in full.layers:
        layer.attn.qkv_proj.bias.copy_(
            torch.randn(layer.attn.qkv_proj.bias.shape, generator=gen))
    ckpt = os.path.join(tmp, "qwen_ckpt")
    if rank == 0:
        save_llama_safetensors(full, ckpt)
    torch.distributed.barrier()

    L, P, D = 9, 16, spec.head_dim

    def run(model, kvh):
        caches = [(torch.zeros(2, kvh, P, D), torch.zeros(2, kvh, P, D))
                  for _ in range(spec.n_layers)]
        meta = AttnMetadata(
            mode="prefill",
            positions=torch.arange(L, dtype=torch.int32),
            slot_mapping=torch.arange(L, dtype=torch.int32),
            block_tables=torch.arange(2, dtype=torch.int32).unsqueeze(0),
            context_lens=torch.tensor([L], dtype=torch.int32),
            cu_seqlens_q=torch.tensor([0, L], dtype=torch.int32),
        )
        h = model(torch.arange(3, 3 + L), caches, meta)
        return model.compute_logits(h)

    full_logits = run(full, spec.n_kv_heads)
    # phase 2: TP=2 model through the loader's qkv-bias shard carving
    pdist.init_distributed(tp_size=world, backend="gloo")
    tp = LlamaForCausalLM(spec, dtype=torch.float32)
    load_llama_safetensors(tp, ckpt)
    tp_logits = run(tp, spec.n_kv_heads // world)
    q.put((rank, (tp_logits - full_logits).abs().max().item()))


@pytest.mark.timeout(300)
def test_tp2_qwen_bias_loader_matches_single_process(tmp_path):
    """Qwen2-family under TP=2: the fused QKV bias must shard with the
    head rows (parallel/tp.py ColumnParallelLinear bias) and reproduce the
    unsharded forward exactly (fp32/gloo)."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29533
    ps = [ctx.Process(target=_qwen_tp_worker, args=(r, 2, port, str(tmp_path), q))
          for r in range(2)]
    [p.start() for p in ps]
    errs = [q.get() for _ in range(2)]
    [p.join(timeout=240) for p in ps]
    for rank, err in errs:
        assert err < 1e-4, (rank, err)
