#!/usr/bin/env python3
"""Flagship benchmark: incidents/hr end-to-end RCA (Llama-3-8B agent).

Driver contract:
    python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run with one rank
per GPU (DP: each rank serves its own shard of concurrent investigations —
weak scaling, the reference's Celery-worker model mapped onto GPUs).

Two modes:

* ``--mode e2e`` (default, the headline): one STEP = --concurrency synthetic
  PagerDuty webhooks driven through the REAL service pipeline on this rank —
  webhook normalise/correlate -> incident row -> queue -> admission ->
  run_background_chat -> the guardrailed agent workflow (input rail, per-turn
  ToolCallGrammar-constrained action decoding, real tool execution against
  the org store / GPU RAG index, per-turn embedding-rail classification of
  tool outputs) -> forced final RCA -> summarization LLM pass ->
  recommendations -> citations -> postmortem row. The timed region contains
  ALL of it (reference pipeline: chat/background/task.py:447-697; metric
  definition BASELINE.json). p50 time-to-RCA = webhook send -> completion
  event per incident.

* ``--mode engine``: the round-1 engine-level path (per-turn RAG + rail
  around raw eng.step() decode turns) — quoted alongside for comparability.

Data: synthetic alert payloads / token streams, random-init weights
(no network in this environment — BASELINE.md).

Rank 0 prints ONE JSON line; value = whole-job incidents/hr over all ranks.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import statistics
import time

import torch


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=2)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--mode", choices=("e2e", "engine"),
                    default=os.environ.get("BENCH_MODE", "e2e"))
    ap.add_argument("--model", default=os.environ.get("BENCH_MODEL", "llama3-8b"))
    ap.add_argument("--concurrency", type=int,
                    default=int(os.environ.get("BENCH_CONCURRENCY", "128")))
    ap.add_argument("--turns", type=int, default=int(os.environ.get("BENCH_TURNS", "4")))
    ap.add_argument("--first-prompt", type=int, default=768)
    ap.add_argument("--turn-prompt", type=int, default=512)
    ap.add_argument("--decode-per-turn", type=int, default=128)
    ap.add_argument("--device", default="auto")
    ap.add_argument("--rag", type=int, default=int(os.environ.get("BENCH_RAG", "1")),
                    help="per-turn on-GPU RAG retrieval + input-rail check (1=on)")
    ap.add_argument("--rag-docs", type=int,
                    default=int(os.environ.get("BENCH_RAG_DOCS", "10000")))
    return ap.parse_args()


def synth_tokens(rng: torch.Generator, n: int, vocab: int) -> list:
    return torch.randint(0, min(vocab - 4, 250), (n,), generator=rng).tolist()


# --------------------------------------------------------------------------
# engine mode (round-1 comparability path)
# --------------------------------------------------------------------------

class RagPath:
    """Engine mode: per-turn retrieval over an HBM-resident corpus
    (on-GPU sentence encoder + kNN) + the input-rail heuristic check."""

    def __init__(self, args, device: str):
        import torch as _t
        from aurora_amd import ops
        from aurora_amd.models.encoder import SentenceEncoder
        self.ops = ops
        self.device = device
        self.encoder = SentenceEncoder(
            "bge-base", dtype=_t.bfloat16 if device != "cpu" else _t.float32,
            device=device)
        self.encoder.init_random(3)
        rng = _t.Generator().manual_seed(17)
        chunks = [_t.randint(1, 250, (96,), generator=rng).tolist()
                  for _ in range(args.rag_docs)]
        embs = []
        for i in range(0, len(chunks), 512):
            embs.append(self.encoder.encode_ids(chunks[i:i + 512]))
        self.docs = _t.cat(embs)
        self.rail_patterns = None

    def rail_check(self, tokens) -> bool:
        from aurora_amd.guardrails.input_rail import _INJECTION_PATTERNS
        text = bytes(t % 256 for t in tokens[:512]).decode("latin-1")
        return not any(p.search(text) for p in _INJECTION_PATTERNS)

    def retrieve(self, query_tokens) -> list:
        qv = self.encoder.encode_ids([query_tokens[:96]])
        vals, idx = self.ops.knn_topk(qv, self.docs, 4)
        out = []
        for i in idx[0].tolist():
            out.extend(((i * 131 + j * 7) % 250 for j in range(32)))
        return out


def run_step_engine(eng, args, rng, vocab, step_idx: int, rag=None):
    """One engine-level investigation batch; returns per-incident latencies."""
    from aurora_amd.engine.sampling_params import SamplingParams

    B, turns = args.concurrency, args.turns
    params = SamplingParams(temperature=0.7, top_p=0.95,
                            max_tokens=args.decode_per_turn, ignore_eos=True)
    ctx_tokens = {i: synth_tokens(rng, args.first_prompt, vocab) for i in range(B)}
    turn_no = {i: 0 for i in range(B)}
    t_start = {i: time.monotonic() for i in range(B)}
    t_done = {}

    def submit(i):
        rid = f"s{step_idx}-inc{i}-t{turn_no[i]}"
        if rag is not None:
            rag.rail_check(ctx_tokens[i][-args.turn_prompt:])
            retrieved = rag.retrieve(ctx_tokens[i][-96:])
            ctx_tokens[i][-len(retrieved):] = retrieved  # fixed token count
        eng.add_request(rid, list(ctx_tokens[i]), params, session_id=f"inc-{step_idx}-{i}")
        return rid

    live = {}
    for i in range(B):
        live[submit(i)] = i
    gen_out = {i: [] for i in range(B)}
    while live:
        for out in eng.step():
            i = live.get(out.request_id)
            if i is None:
                continue
            gen_out[i].extend(out.new_token_ids)
            if out.finished:
                del live[out.request_id]
                ctx_tokens[i].extend(gen_out[i][-args.decode_per_turn:])
                turn_no[i] += 1
                if turn_no[i] < turns:
                    ctx_tokens[i].extend(synth_tokens(rng, args.turn_prompt, vocab))
                    live[submit(i)] = i
                else:
                    t_done[i] = time.monotonic()
    return [t_done[i] - t_start[i] for i in range(B)]


def main_engine(args, rank, world, use_cuda):
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.models.config import get_spec
    from aurora_amd.parallel import dist as pdist

    spec = get_spec(args.model)
    cfg = EngineConfig(
        model=args.model,
        device="cpu" if not use_cuda else "auto",
        max_num_seqs=max(args.concurrency + 2, 8),
        max_batched_tokens=16384,
        seed=1234 + rank,
    )
    eng = LLMEngine(cfg)
    rng = torch.Generator().manual_seed(999 + rank)
    vocab = spec.vocab_size
    rag = RagPath(args, eng.device) if args.rag else None

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
        if world > 1:
            pdist.barrier()
            if use_cuda:
                torch.cuda.synchronize()

    for w in range(args.warmup):
        run_step_engine(eng, args, rng, vocab, -1 - w, rag)
    sync()
    t0 = time.monotonic()
    lat = []
    for s in range(args.steps):
        lat.extend(run_step_engine(eng, args, rng, vocab, s, rag))
    sync()
    elapsed = time.monotonic() - t0
    extra = {
        "pipeline": "engine",
        "rag": bool(rag), "rag_docs": args.rag_docs if rag else 0,
        "guardrail_per_turn": bool(rag),
        "decode_tokens_per_turn": args.decode_per_turn,
        "prompt_tokens_per_incident": args.first_prompt
        + (args.turns - 1) * args.turn_prompt,
        "engine_timers_s": {k: round(v, 2) for k, v in eng.timers.items()},
        "engine_counters": eng.counters,
    }
    return elapsed, lat, extra


# --------------------------------------------------------------------------
# e2e mode (the headline): real webhook -> worker -> workflow -> summary
# --------------------------------------------------------------------------

#: fast read-only tools the random-weight agent may call during the bench —
#: real registry tools hitting the org store / infra graph / GPU RAG
#: ("stub-fast" tool execution per the metric definition; exec-style tools
#: are excluded so a random-weight model can't run shell commands)
BENCH_TOOLS = [
    "knowledge_base_search", "list_incidents", "get_incident",
    "get_incident_timeline", "get_alert_field", "list_services",
    "get_service_impact", "get_infrastructure_context", "final_answer",
]

_LOREM = ("deploy rollout canary latency p99 spike postgres connection pool "
          "exhausted kafka lag consumer rebalance oom killed node pressure "
          "dns resolution timeout ingress 502 upstream retry storm cache "
          "eviction hot shard gc pause tls handshake certificate expired").split()


def _synth_doc(rng: torch.Generator, n_words: int = 160) -> str:
    idx = torch.randint(0, len(_LOREM), (n_words,), generator=rng).tolist()
    return " ".join(_LOREM[i] for i in idx)


def _build_runtime(args, rank, use_cuda):
    os.environ.setdefault("AURORA_RAIL_BACKEND", "encoder")
    from aurora_amd.engine.engine import EngineConfig
    from aurora_amd.engine.server import AsyncEngineServer
    from aurora_amd.service.runtime import Runtime
    from aurora_amd.utils.config import Settings

    scfg = Settings(db_path=":memory:",
                    secrets_path=f"/tmp/bench_secrets_{rank}.json")
    scfg.worker_concurrency = args.concurrency + 8
    scfg.max_concurrent_rca = args.concurrency * 2
    scfg.rca_max_turns = args.turns
    scfg.guardrails_enabled = True
    ecfg = EngineConfig(
        model=args.model,
        device="cpu" if not use_cuda else "auto",
        max_num_seqs=max(args.concurrency + 2, 8),
        max_batched_tokens=16384,
        kv_memory_fraction=float(os.environ.get("KV_MEMORY_FRACTION", "0.8")),
        seed=1234 + rank,
    )
    server = AsyncEngineServer(ecfg)
    rt = Runtime(scfg, engine_server=server)
    org_id = rt.admin.create_org(f"bench-{rank}")
    rt.admin.create_user(org_id, f"bench{rank}@aurora.local",
                         user_id=f"bench-user-{rank}")
    # infra graph for the graph tools
    g = rt.graph(org_id)
    for svc in ("api", "db", "cache", "queue", "ingress"):
        g.upsert_service(svc)
    for a, b in (("api", "db"), ("api", "cache"), ("ingress", "api"),
                 ("queue", "db")):
        g.add_dependency(a, b)
    # RAG corpus: args.rag_docs synthetic chunks embedded on the GPU encoder
    # (HBM-resident kNN bank) without round-tripping SQLite at bench setup
    rag = rt.rag(org_id)
    rng = torch.Generator().manual_seed(17)
    docs = [_synth_doc(rng) for _ in range(args.rag_docs)]
    BATCH = 512
    for i in range(0, len(docs), BATCH):
        batch = docs[i:i + BATCH]
        vecs = rag._embed(batch)
        for j, d in enumerate(batch):
            rag.chunks.append({"text": d, "doc_id": f"kb-{i + j}",
                               "doc_name": f"runbook-{(i + j) % 97}"})
            rag.bm25.add(d)
        if vecs is not None:
            vecs = vecs.to(rag.device)
            rag.emb = vecs if rag.emb is None else torch.cat([rag.emb, vecs])
    return rt, org_id


def _payload(rng: torch.Generator, step: int, i: int) -> dict:
    """Unique synthetic PagerDuty v3 event. Every alert must open a NEW
    incident, so its text is made lexically unique (the correlator's
    similarity strategy would otherwise merge same-vocabulary alerts —
    which is correct behavior for real alert storms)."""
    tag = f"s{step}i{i}"
    doc = " ".join(f"{w}{tag}" for w in _synth_doc(rng, 60).split())
    return {"event": {"data": {
        "id": f"PD-{step}-{i}-{torch.randint(0, 1 << 30, (1,), generator=rng).item()}",
        "title": f"[bench s{step}] {doc[:80]} #{i}",
        "service": {"summary": f"svc-{step}-{i}"},
        "priority": {"summary": "P2"},
        "urgency": "high",
        "body": {"details": doc},
    }}}


async def run_step_e2e(rt, org_id, args, rng, step: int,
                       deadline_s: float = 1500.0):
    import sys
    from aurora_amd.service.webhooks import process_webhook
    B = args.concurrency
    ev_q = rt.events.subscribe(org_id)
    t_start: dict = {}
    pending = set()
    statuses: dict = {}
    t0 = time.monotonic()
    for i in range(B):
        t_start_i = time.monotonic()
        iid, created = process_webhook(
            rt, org_id, "pagerduty", _payload(rng, step, i),
            rca_countdown_s=0.0,
            rca_args={"max_turns": args.turns, "tool_subset": BENCH_TOOLS})
        assert created, f"correlator merged bench alert {i} into {iid}"
        t_start[iid] = t_start_i
        pending.add(iid)
    lat = []
    last_beat = t0
    while pending:
        left = deadline_s - (time.monotonic() - t0)
        if left <= 0:
            # stragglers the deadline cut off: reported, and EXCLUDED from
            # the throughput numerator (work the timed region didn't finish)
            print(f"[bench] step {step}: deadline, {len(pending)} pending",
                  file=sys.stderr, flush=True)
            statuses["straggler"] = statuses.get("straggler", 0) + len(pending)
            break
        try:
            ev = await asyncio.wait_for(ev_q.get(), timeout=min(left, 30.0))
        except asyncio.TimeoutError:
            ev = None
        now = time.monotonic()
        if now - last_beat > 30.0:
            last_beat = now
            c = rt.engine.engine.counters
            print(f"[bench] step {step} t={now - t0:.0f}s done="
                  f"{len(lat)}/{B} counters={c}", file=sys.stderr, flush=True)
        if (ev and ev.get("type") == "incident_update"
                and ev.get("incident_id") in pending
                and ev.get("aurora_status") in ("completed", "failed")):
            iid = ev["incident_id"]
            pending.discard(iid)
            statuses[ev["aurora_status"]] = statuses.get(ev["aurora_status"], 0) + 1
            lat.append(time.monotonic() - t_start[iid])
    rt.events.unsubscribe(org_id, ev_q)
    return lat, statuses


async def main_e2e_async(args, rank, world, use_cuda):
    from aurora_amd.parallel import dist as pdist

    rt, org_id = _build_runtime(args, rank, use_cuda)
    rt.start()
    rng = torch.Generator().manual_seed(999 + rank)

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
        if world > 1:
            pdist.barrier()
            if use_cuda:
                torch.cuda.synchronize()

    statuses_all: dict = {}
    for w in range(args.warmup):
        await run_step_e2e(rt, org_id, args, rng, -1 - w)
    sync()
    t0 = time.monotonic()
    lat = []
    for s in range(args.steps):
        l, st = await run_step_e2e(rt, org_id, args, rng, s)
        lat.extend(l)
        for k, v in st.items():
            statuses_all[k] = statuses_all.get(k, 0) + v
    sync()
    elapsed = time.monotonic() - t0

    eng = rt.engine.engine
    extra = {
        "pipeline": "webhook->queue->workflow->summary (end-to-end)",
        "in_timed_region": ["webhook normalise+correlate", "queue dispatch",
                           "input rail", "per-turn grammar tool decode",
                           "tool execution", "per-turn tool-output rail",
                           "GPU RAG (kb tool)", "forced final RCA",
                           "summarization pass", "recommendations",
                           "citations", "postmortem"],
        "rag_docs": args.rag_docs,
        "guardrail_per_turn": True,
        "tool_subset": BENCH_TOOLS,
        "completed": statuses_all.get("completed", 0),
        "failed_terminal": statuses_all.get("failed", 0),
        "stragglers": statuses_all.get("straggler", 0),
        "engine_timers_s": {k: round(v, 2) for k, v in eng.timers.items()},
        "engine_counters": eng.counters,
    }
    await rt.stop()
    rt.engine.shutdown()
    return elapsed, lat, extra


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    use_cuda = torch.cuda.is_available() and args.device != "cpu"

    from aurora_amd.parallel import dist as pdist
    if world > 1:
        pdist.init_distributed(tp_size=1)

    if args.mode == "engine":
        elapsed, lat, extra = main_engine(args, rank, world, use_cuda)
    else:
        elapsed, lat, extra = asyncio.run(
            main_e2e_async(args, rank, world, use_cuda))

    # incidents that reached a terminal status inside the deadline; in
    # engine mode every submitted incident completes by construction.
    # Stragglers cut off by the e2e deadline must NOT count in the
    # numerator — that would credit work the timed region didn't finish.
    done_local = float(extra.get("completed", 0) + extra.get("failed_terminal", 0)
                       if args.mode == "e2e"
                       else args.steps * args.concurrency)

    # max over ranks (slowest rank defines the job); sum of terminal counts
    if world > 1:
        import torch.distributed as dist
        dev = "cuda" if (use_cuda and dist.get_backend() == "nccl") else "cpu"
        t = torch.tensor([elapsed], device=dev)  # RCCL needs device tensors
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])
        d = torch.tensor([done_local], device=dev)
        dist.all_reduce(d, op=dist.ReduceOp.SUM)
        incidents = int(d[0])
    else:
        incidents = int(done_local)
    value = incidents / (elapsed / 3600.0)
    p50 = statistics.median(lat) if lat else 0.0
    tokens_per_incident = (args.first_prompt + (args.turns - 1) * args.turn_prompt
                           + args.turns * args.decode_per_turn)
    if rank == 0:
        print(json.dumps({
            "metric": "incidents/hr",
            "value": round(value, 2),
            "unit": "incidents/hr",
            "n_gpus": world if use_cuda else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "mode": args.mode,
                "global_batch": args.concurrency * world,
                "seq_len": tokens_per_incident,
                "parallelism": f"dp{world}",
                "concurrency_per_gpu": args.concurrency,
                "turns": args.turns,
                "p50_time_to_rca_s": round(p50, 3),
                **extra,
            },
        }))


if __name__ == "__main__":
    main()
