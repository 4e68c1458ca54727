"""GPU numerics tests: every gfx950 HIP kernel vs the plain-PyTorch fp32
reference of the same op (aurora_amd.ops.torch_ref), on the same bf16 inputs.

All marked @pytest.mark.gpu (run via gpurun on a real MI355X)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from aurora_amd import ops
from aurora_amd.ops import torch_ref as R


def _err(a, b):
    a = a.float().cpu()
    b = b.float().cpu()
    denom = b.abs().max().clamp(min=1e-6)
    return ((a - b).abs().max() / denom).item()


@pytest.fixture(autouse=True)
def _require_ext():
    assert ops.hip_available(), "HIP extension must be built in-tree"


def test_rmsnorm():
    x = torch.randn(37, 4096, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(4096, device="cuda", dtype=torch.bfloat16)
    y = ops.rmsnorm(x, w, 1e-5)
    ref = R.rmsnorm(x.cpu(), w.cpu(), 1e-5)
    assert _err(y, ref) < 2e-2


def test_rmsnorm_residual():
    x = torch.randn(16, 512, device="cuda", dtype=torch.bfloat16)
    r = torch.randn(16, 512, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(512, device="cuda", dtype=torch.bfloat16)
    y, nr = ops.rmsnorm_residual(x, r, w, 1e-5)
    ry, rnr = R.rmsnorm_residual(x.cpu(), r.cpu(), w.cpu(), 1e-5)
    assert _err(nr, rnr) < 1e-2
    assert _err(y, ry) < 2e-2


def test_rope():
    T, Hq, Hkv, D = 33, 8, 2, 128
    cs = R.rope_cos_sin_cache(D, 4096, 500000.0)
    q = torch.randn(T, Hq, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(T, Hkv, D, device="cuda", dtype=torch.bfloat16)
    pos = torch.randint(0, 4096, (T,), dtype=torch.int32, device="cuda")
    q0, k0 = q.cpu(), k.cpu()
    q2, k2 = ops.rope_apply(q, k, pos, cs.cuda())
    rq, rk = R.rope_apply(q0, k0, pos.cpu(), cs)
    assert _err(q2, rq) < 2e-2
    assert _err(k2, rk) < 2e-2


def test_silu_mul():
    x = torch.randn(64, 2048, device="cuda", dtype=torch.bfloat16)
    y = ops.silu_mul(x)
    assert _err(y, R.silu_mul(x.cpu())) < 2e-2


def test_kv_cache_write():
    T, Hkv, D, P, NP = 50, 4, 128, 64, 8
    k = torch.randn(T, Hkv, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(T, Hkv, D, device="cuda", dtype=torch.bfloat16)
    kc = torch.zeros(NP, Hkv, P, D, device="cuda", dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    perm = torch.randperm(NP * P)[:T].to(torch.int32)
    ops.kv_cache_write(k, v, kc, vc, perm.cuda())
    kc_ref = torch.zeros(NP, Hkv, P, D)
    vc_ref = torch.zeros_like(kc_ref)
    R.kv_cache_write(k.cpu().float(), v.cpu().float(), kc_ref, vc_ref, perm)
    assert _err(kc, kc_ref) < 1e-2
    assert _err(vc, vc_ref) < 1e-2


@pytest.mark.parametrize("D,G,Hkv", [(128, 4, 8), (128, 1, 4), (64, 8, 2), (128, 8, 1)])
def test_paged_decode(D, G, Hkv):
    torch.manual_seed(0)
    B, P, NP = 5, 64, 40
    Hq = G * Hkv
    ctx = torch.tensor([17, 64, 130, 1, 301], dtype=torch.int32)
    maxp = (int(ctx.max()) + P - 1) // P
    bt = torch.zeros(B, maxp, dtype=torch.int32)
    used = torch.randperm(NP)
    u = 0
    for b in range(B):
        need = (int(ctx[b]) + P - 1) // P
        bt[b, :need] = used[u:u + need].to(torch.int32)
        u += need
    kc = torch.randn(NP, Hkv, P, D, device="cuda", dtype=torch.bfloat16)
    vc = torch.randn(NP, Hkv, P, D, device="cuda", dtype=torch.bfloat16)
    q = torch.randn(B, Hq, D, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(D)
    out = ops.paged_attention_decode(q, kc, vc, bt.cuda(), ctx.cuda(), scale)
    ref = R.paged_attention_decode(q.cpu(), kc.cpu(), vc.cpu(), bt, ctx, scale)
    assert _err(out, ref) < 3e-2


@pytest.mark.parametrize("D,G,Hkv", [(128, 4, 2), (64, 1, 4)])
def test_prefill_paged_fresh(D, G, Hkv):
    """Fresh prefill (qlen == ctx), ragged lengths crossing 128-row blocks."""
    torch.manual_seed(1)
    P, NP = 256, 24
    Hq = G * Hkv
    lens = [5, 200, 513, 128]
    T = sum(lens)
    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)), dtype=torch.int32)
    ctx = torch.tensor(lens, dtype=torch.int32)
    maxp = (max(lens) + P - 1) // P
    bt = torch.zeros(len(lens), maxp, dtype=torch.int32)
    used = torch.randperm(NP)
    u = 0
    slot_map = torch.empty(T, dtype=torch.int32)
    for s, L in enumerate(lens):
        need = (L + P - 1) // P
        pages = used[u:u + need]
        bt[s, :need] = pages.to(torch.int32)
        u += need
        a = int(cu[s])
        for i in range(L):
            slot_map[a + i] = int(pages[i // P]) * P + i % P
    q = torch.randn(T, Hq, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(T, Hkv, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(T, Hkv, D, device="cuda", dtype=torch.bfloat16)
    kc = torch.zeros(NP, Hkv, P, D, device="cuda", dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    ops.kv_cache_write(k, v, kc, vc, slot_map.cuda())
    scale = 1.0 / math.sqrt(D)
    out = ops.attention_prefill_paged(q, kc, vc, bt.cuda(), cu.cuda(), ctx.cuda(), scale)
    ref = R.attention_prefill_paged(q.cpu(), kc.cpu(), vc.cpu(), bt, cu, ctx, scale)
    assert _err(out, ref) < 3e-2


def test_prefill_paged_append():
    """Append-prefill: 100 cached tokens + 37 new queries."""
    torch.manual_seed(2)
    D, Hq, Hkv, P, NP = 128, 8, 2, 256, 8
    hist, new = 100, 37
    L = hist + new
    pages = [3, 1]
    bt = torch.tensor([pages], dtype=torch.int32)
    slot_map = torch.tensor([pages[i // P] * P + i % P for i in range(L)], dtype=torch.int32)
    k = torch.randn(L, Hkv, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(L, Hkv, D, device="cuda", dtype=torch.bfloat16)
    kc = torch.zeros(NP, Hkv, P, D, device="cuda", dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    ops.kv_cache_write(k, v, kc, vc, slot_map.cuda())
    q = torch.randn(new, Hq, D, device="cuda", dtype=torch.bfloat16)
    cu = torch.tensor([0, new], dtype=torch.int32)
    ctx = torch.tensor([L], dtype=torch.int32)
    scale = 1.0 / math.sqrt(D)
    out = ops.attention_prefill_paged(q, kc, vc, bt.cuda(), cu.cuda(), ctx.cuda(), scale)
    ref = R.attention_prefill_paged(q.cpu(), kc.cpu(), vc.cpu(), bt, cu, ctx, scale)
    assert _err(out, ref) < 3e-2


@pytest.mark.parametrize("causal", [False, True])
def test_varlen_attention_encoder(causal):
    torch.manual_seed(3)
    D, H = 64, 12
    lens = [7, 128, 33]
    T = sum(lens)
    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)), dtype=torch.int32)
    q = torch.randn(T, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(T, H, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(T, H, D, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(D)
    out = ops.attention_prefill(q, k, v, cu.cuda(), scale, causal)
    ref = R.attention_prefill(q.cpu(), k.cpu(), v.cpu(), cu, scale, causal)
    assert _err(out, ref) < 3e-2


def test_embed_pool():
    h = torch.randn(100, 768, device="cuda", dtype=torch.bfloat16)
    cu = torch.tensor([0, 10, 100], dtype=torch.int32)
    out = ops.embed_pool(h, cu.cuda())
    ref = R.embed_pool(h.cpu(), cu)
    assert _err(out, ref) < 2e-2


def test_knn():
    docs = torch.nn.functional.normalize(
        torch.randn(5000, 768), dim=-1).to("cuda", torch.bfloat16)
    qv = docs[123:125].float() + 0.001 * torch.randn(2, 768, device="cuda")
    vals, idx = ops.knn_topk(qv.to(torch.bfloat16), docs, 10)
    assert idx[0, 0].item() == 123 and idx[1, 0].item() == 124


def test_engine_gpu_matches_cpu_reference():
    """Whole-engine check: GPU bf16 forward vs CPU fp32 same-weight model."""
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    prompts = [list(range(10, 200)), list(range(300, 350)), [7, 8, 9]]
    p = SamplingParams(temperature=0.0, max_tokens=16, ignore_eos=True)
    gpu = LLMEngine(EngineConfig(model="smoke-llama", device="cuda", seed=11,
                                 kv_cache_pages=64))
    outs_gpu = gpu.generate(prompts, p)
    cpu = LLMEngine(EngineConfig(model="smoke-llama", device="cpu", seed=11))
    outs_cpu = cpu.generate(prompts, p)
    # bf16 vs fp32 on random weights: token-level agreement is statistical;
    # compare the prefill logits directly instead of token ids
    import torch as _t
    ids = _t.tensor(prompts[0], device="cuda")
    from aurora_amd.models.llama import AttnMetadata
    L = len(prompts[0])
    meta = AttnMetadata(
        mode="prefill",
        positions=_t.arange(L, dtype=_t.int32, device="cuda"),
        slot_mapping=_t.arange(L, dtype=_t.int32, device="cuda"),
        block_tables=_t.arange(1, dtype=_t.int32, device="cuda").unsqueeze(0),
        context_lens=_t.tensor([L], dtype=_t.int32, device="cuda"),
        cu_seqlens_q=_t.tensor([0, L], dtype=_t.int32, device="cuda"),
    )
    kv = [(c[0].clone(), c[1].clone()) for c in gpu.kv.caches]
    hidden = gpu.model(ids, kv, meta)
    logits = gpu.model.compute_logits(hidden)
    meta_cpu = AttnMetadata(
        mode="prefill",
        positions=_t.arange(L, dtype=_t.int32),
        slot_mapping=_t.arange(L, dtype=_t.int32),
        block_tables=_t.arange(16, dtype=_t.int32).unsqueeze(0),
        context_lens=_t.tensor([L], dtype=_t.int32),
        cu_seqlens_q=_t.tensor([0, L], dtype=_t.int32),
    )
    P = cpu.kv.page_size
    npage = (L + P - 1) // P
    meta_cpu.block_tables = _t.arange(npage, dtype=_t.int32).unsqueeze(0)
    kv_cpu = [(c[0].clone(), c[1].clone()) for c in cpu.kv.caches]
    hidden_cpu = cpu.model(_t.tensor(prompts[0]), kv_cpu, meta_cpu)
    logits_cpu = cpu.model.compute_logits(hidden_cpu)
    a = logits[-1].float().cpu()
    b = logits_cpu[-1].float()
    cos = torch.nn.functional.cosine_similarity(a, b, dim=0)
    assert cos.item() > 0.98, cos.item()
    # GPU generation must be deterministic under greedy
    gpu2 = LLMEngine(EngineConfig(model="smoke-llama", device="cuda", seed=11,
                                  kv_cache_pages=64))
    assert gpu2.generate(prompts, p) == outs_gpu


def test_engine_gpu_grammar_json():
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.grammar import JsonGrammar, validate_json_bytes
    from aurora_amd.engine.sampling_params import SamplingParams

    eng = LLMEngine(EngineConfig(model="smoke-llama", device="cuda", seed=5,
                                 kv_cache_pages=64))
    p = SamplingParams(temperature=0.8, max_tokens=200,
                       grammar=JsonGrammar(soft_close_bytes=140))
    out = eng.generate([[1, 2, 3]], p)[0]
    data = bytes(t for t in out if t < 256)
    assert validate_json_bytes(data), data


@pytest.mark.parametrize("M,N,K", [(1, 6144, 4096), (16, 6144, 4096),
                                   (16, 128256, 4096), (33, 4096, 14336),
                                   (64, 28672, 4096), (5, 512, 512)])
def test_skinny_gemm(M, N, K):
    from aurora_amd import _hip_ops
    torch.manual_seed(0)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) / (K ** 0.25)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) / (K ** 0.25)
    out = _hip_ops.skinny_gemm(x, w)
    ref = (x.float() @ w.float().t())
    err = (out.float() - ref).abs().max() / ref.abs().max().clamp(min=1e-6)
    assert err.item() < 3e-2, err.item()


@pytest.mark.parametrize("M,N,K,st", [(1, 6144, 4096, 256), (17, 4096, 4096, 256),
                                      (64, 6144, 4096, 512), (64, 4096, 14336, 512),
                                      (64, 28672, 4096, 1), (33, 4000, 1152, 512),
                                      (128, 6144, 4096, 512), (100, 4096, 4096, 512),
                                      (256, 6144, 4096, 512), (256, 4096, 14336, 512),
                                      (200, 28672, 4096, 1), (256, 4000, 1152, 512)])
def test_skinny_gemm2(M, N, K, st):
    """LDS-staged v2 (split and no-split paths, ragged N) vs fp32 torch."""
    from aurora_amd import _hip_ops
    torch.manual_seed(1)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) / (K ** 0.25)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) / (K ** 0.25)
    out = _hip_ops.skinny_gemm2_ex(x, w, st)
    ref = (x.float() @ w.float().t())
    err = (out.float() - ref).abs().max() / ref.abs().max().clamp(min=1e-6)
    assert err.item() < 3e-2, err.item()


def test_fused_rope_kv_write():
    torch.manual_seed(4)
    T, Hq, Hkv, D, P, NP = 29, 8, 2, 128, 64, 4
    cs = R.rope_cos_sin_cache(D, 2048, 500000.0).cuda()
    qkv = torch.randn(T, (Hq + 2 * Hkv) * D, device="cuda", dtype=torch.bfloat16)
    kc = torch.zeros(NP, Hkv, P, D, device="cuda", dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    pos = torch.randint(0, 2048, (T,), dtype=torch.int32, device="cuda")
    slots = torch.randperm(NP * P)[:T].to(torch.int32).cuda()
    q = ops.fused_rope_kv_write(qkv, pos, slots, cs, kc, vc, Hq, Hkv)
    # CPU composed reference on the same inputs
    kc_ref = torch.zeros(NP, Hkv, P, D)
    vc_ref = torch.zeros_like(kc_ref)
    q_ref = ops.fused_rope_kv_write(qkv.cpu().float(), pos.cpu(), slots.cpu(),
                                    cs.cpu(), kc_ref, vc_ref, Hq, Hkv)
    assert _err(q, q_ref) < 2e-2
    assert _err(kc, kc_ref) < 2e-2
    assert _err(vc, vc_ref) < 2e-2


def test_engine_gpu_moe():
    """Mixtral-style MoE through the GPU engine (HIP kernels + routing)."""
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams
    eng = LLMEngine(EngineConfig(model="smoke-moe", device="cuda", seed=9,
                                 kv_cache_pages=64))
    p = SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True)
    outs = eng.generate([[5, 6, 7, 8] * 10, [9] * 17], p)
    assert all(len(o) == 8 for o in outs)
    eng2 = LLMEngine(EngineConfig(model="smoke-moe", device="cuda", seed=9,
                                  kv_cache_pages=64))
    assert eng2.generate([[5, 6, 7, 8] * 10, [9] * 17], p) == outs


def test_engine_gpu_multi_model_cohost():
    """Two in-process engines co-host on one GPU (per-purpose aux models,
    service/runtime.py engine_for): both must generate."""
    import asyncio
    from aurora_amd.engine.engine import EngineConfig
    from aurora_amd.engine.server import AsyncEngineServer
    from aurora_amd.engine.sampling_params import SamplingParams
    from aurora_amd.service.runtime import Runtime
    from aurora_amd.utils.config import Settings
    server = AsyncEngineServer(EngineConfig(
        model="smoke-llama", device="cuda", seed=1, kv_memory_fraction=0.05,
        use_hip_graphs=False))
    rt = Runtime(Settings(db_path=":memory:", secrets_path="/tmp/mm_sec.json",
                          main_model="smoke-llama",
                          summarization_model="smoke-moe",
                          engine_device="cuda"),
                 engine_server=server)
    try:
        org = rt.admin.create_org("mm")
        llm = rt.llm(org)
        p = SamplingParams(temperature=0.0, max_tokens=4, ignore_eos=True)

        async def go():
            a = await llm.invoke("alert: db down", p, purpose="main")
            b = await llm.invoke("summarize: db down", p, purpose="summarization")
            return a, b
        a, b = asyncio.run(go())
        assert len(a.token_ids) == 4 and len(b.token_ids) == 4, (a, b)
        assert str(rt._aux_engines["smoke-moe"].engine.device).startswith("cuda")
    finally:
        asyncio.run(rt.stop())
        server.shutdown()


def test_moe_grouped_gemm_numerics():
    """Packed grouped GEMM vs per-expert fp32 matmul, incl. empty experts
    and uneven counts."""
    from aurora_amd import _hip_ops
    torch.manual_seed(3)
    E, K, N = 8, 512, 256
    counts = [0, 5, 256, 1, 0, 33, 17, 64]
    A = sum(counts)
    starts = [0]
    for c in counts[:-1]:
        starts.append(starts[-1] + c)
    xp = torch.randn(A, K, device="cuda", dtype=torch.bfloat16) / (K ** 0.25)
    w = torch.randn(E, N, K, device="cuda", dtype=torch.bfloat16) / (K ** 0.25)
    out = _hip_ops.moe_grouped_gemm(
        xp, w, torch.tensor(starts, dtype=torch.int32, device="cuda"),
        torch.tensor(counts, dtype=torch.int32, device="cuda"))
    ref = torch.empty(A, N, device="cuda", dtype=torch.float32)
    for e in range(E):
        s, c = starts[e], counts[e]
        if c:
            ref[s:s + c] = xp[s:s + c].float() @ w[e].float().t()
    err = (out.float() - ref).abs().max().item() / max(
        ref.abs().max().item(), 1e-6)
    assert err < 2e-2, err


def test_moe_mlp_grouped_vs_bmm_path():
    """The packed grouped decode path and the round-1 bmm fallback agree."""
    import os
    from aurora_amd.models.config import get_spec
    from aurora_amd.models.llama import MoEMLP
    spec = get_spec("smoke-moe")
    torch.manual_seed(5)
    m = MoEMLP(spec, torch.bfloat16).cuda()
    for p in m.parameters():
        p.data.normal_(0, spec.hidden_size ** -0.5)
    x = torch.randn(64, spec.hidden_size, device="cuda",
                    dtype=torch.bfloat16)
    y_grouped = m(x).float()
    os.environ["AURORA_MOE_GROUPED"] = "0"
    try:
        m._w_gu = m._w_dn = None
        y_bmm = m(x).float()
    finally:
        os.environ.pop("AURORA_MOE_GROUPED")
    err = (y_grouped - y_bmm).abs().max().item() / max(
        y_bmm.abs().max().item(), 1e-6)
    assert err < 2e-2, err


# -------------------------------------------------- long-context (32k-128k)


@pytest.mark.parametrize("ctx_len", [32768, 131072])
def test_paged_decode_long_context(ctx_len):
    """Decode numerics at 32k/128k contexts: the flash-decode context
    split and multi-page block tables under real lengths (VERDICT r1
    item 9; round-1 GPU tests stopped at 513 tokens)."""
    torch.manual_seed(2)
    D, Hkv, G = 128, 2, 4
    P = 256
    Hq = G * Hkv
    B = 2
    lens = [ctx_len, ctx_len // 2 + 77]
    NP = sum((l + P - 1) // P for l in lens) + 2
    ctx = torch.tensor(lens, dtype=torch.int32)
    maxp = (max(lens) + P - 1) // P
    bt = torch.zeros(B, maxp, dtype=torch.int32)
    used = torch.randperm(NP)
    u = 0
    for b, L in enumerate(lens):
        need = (L + P - 1) // P
        bt[b, :need] = used[u:u + need].to(torch.int32)
        u += need
    kc = torch.randn(NP, Hkv, P, D, device="cuda", dtype=torch.bfloat16) / 4
    vc = torch.randn(NP, Hkv, P, D, device="cuda", dtype=torch.bfloat16) / 4
    q = torch.randn(B, Hq, D, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(D)
    out = ops.paged_attention_decode(q, kc, vc, bt.cuda(), ctx.cuda(), scale)
    # reference: gather each seq's KV from the page table, fp32 attention
    # (cheap at decode: one query row per head)
    for b, L in enumerate(lens):
        pages = bt[b, : (L + P - 1) // P].long()
        k = kc[pages].permute(1, 0, 2, 3).reshape(Hkv, -1, D)[:, :L].float().cpu()
        v = vc[pages].permute(1, 0, 2, 3).reshape(Hkv, -1, D)[:, :L].float().cpu()
        qb = q[b].float().cpu()  # [Hq, D]
        for h in range(Hq):
            kh = k[h // G]
            att = (qb[h] @ kh.t()) * scale
            w = att.softmax(-1)
            ref = w @ v[h // G]
            got = out[b, h].float().cpu()
            err = (got - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
            assert err < 3e-2, (b, h, err)


def test_engine_long_context_prefill_decode_evict():
    """Engine integrity at 32k+ real context: chunked prefill across many
    pages, decode with the long block table, then KV-pressure eviction
    (session KV evict/restore) — all on the smoke model."""
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    eng = LLMEngine(EngineConfig(model="smoke-llama-64k", device="cuda", seed=9,
                                 max_model_len=40960, kv_cache_pages=200,
                                 max_batched_tokens=8192))
    P = eng.kv.page_size
    long_prompt = torch.randint(1, 900, (33000,)).tolist()
    eng.add_request("long-1", long_prompt,
                    SamplingParams(max_tokens=8, ignore_eos=True),
                    session_id="sess-long")
    outs = []
    for _ in range(40):
        outs.extend(eng.step())
        if not eng.has_work():
            break
    assert outs and outs[-1].finished
    assert sum(len(o.new_token_ids) for o in outs) == 8
    # the finished seq persists as a session with a long page chain
    sess = eng.sessions["sess-long"]
    assert len(sess.pages) >= 33000 // P
    # KV pressure: a second long request must evict the session cleanly
    eng.add_request("long-2", torch.randint(1, 900, (30000,)).tolist(),
                    SamplingParams(max_tokens=4, ignore_eos=True))
    for _ in range(40):
        eng.step()
        if not eng.has_work():
            break
    assert "sess-long" not in eng.sessions  # evicted under pressure
    assert eng.kv.num_free() > 0


# ----------------------------------------------------------- fp8 KV cache


def test_fp8_kv_roundtrip_and_decode():
    """e4m3 KV cache: write kernel converts bf16->fp8, decode reads fp8 —
    numerics vs the fp32 reference within the documented fp8 tolerance
    (docs/ROUND2.md §3)."""
    from aurora_amd import _hip_ops
    torch.manual_seed(4)
    D, Hkv, G, P, NP, B = 128, 2, 4, 64, 12, 3
    Hq = G * Hkv
    lens = [17, 130, 301]
    # build both caches through the same write kernel
    kc8 = torch.zeros(NP, Hkv, P, D, device="cuda",
                      dtype=torch.float8_e4m3fn)
    vc8 = torch.zeros_like(kc8)
    kc16 = torch.zeros(NP, Hkv, P, D, device="cuda", dtype=torch.bfloat16)
    vc16 = torch.zeros_like(kc16)
    bt = torch.zeros(B, (max(lens) + P - 1) // P, dtype=torch.int32)
    used = torch.randperm(NP)
    u = 0
    slot_rows = []
    kv_rows = []
    for b, L in enumerate(lens):
        need = (L + P - 1) // P
        pages = used[u:u + need]
        bt[b, :need] = pages.to(torch.int32)
        u += need
        for t in range(L):
            slot_rows.append(int(pages[t // P]) * P + t % P)
    T = len(slot_rows)
    k = torch.randn(T, Hkv, D, device="cuda", dtype=torch.bfloat16) / 4
    v = torch.randn(T, Hkv, D, device="cuda", dtype=torch.bfloat16) / 4
    slots = torch.tensor(slot_rows, dtype=torch.int32, device="cuda")
    _hip_ops.kv_cache_write(k, v, kc8, vc8, slots)
    _hip_ops.kv_cache_write(k, v, kc16, vc16, slots)
    # round-trip error of the fp8 cache itself: e4m3 has a 2^-3 relative
    # step, so abs error near |x|~1.5 (randn/4 tail) is up to 0.0625
    rt = kc8.float() - kc16.float()
    assert rt.abs().max().item() < 0.07
    q = torch.randn(B, Hq, D, device="cuda", dtype=torch.bfloat16)
    ctx = torch.tensor(lens, dtype=torch.int32)
    scale = 1.0 / math.sqrt(D)
    out8 = ops.paged_attention_decode(q, kc8, vc8, bt.cuda(), ctx.cuda(), scale)
    ref = R.paged_attention_decode(q.cpu(), kc16.cpu(), vc16.cpu(), bt, ctx,
                                   scale)
    # fp8 e4m3: ~2^-3 relative element error, averaged down by the dot
    assert _err(out8, ref) < 8e-2


def test_engine_fp8_kv_end_to_end(monkeypatch):
    """Engine with AURORA_KV_FP8=1: page budget uses 1 byte/elem, chunked
    prefill reads the fp8 cache, decode completes."""
    monkeypatch.setenv("AURORA_KV_FP8", "1")
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams
    eng = LLMEngine(EngineConfig(model="smoke-llama", device="cuda", seed=6,
                                 kv_cache_pages=64, max_batched_tokens=512))
    assert eng.kv.k_caches[0].dtype == torch.float8_e4m3fn
    prompts = [torch.randint(1, 900, (700,)).tolist() for _ in range(3)]
    outs = eng.generate(prompts, SamplingParams(max_tokens=6, ignore_eos=True))
    assert all(len(o) == 6 for o in outs)


def test_engine_gpu_fused_react_turn():
    """The fused thought+action grammar on the GPU engine: every output
    parses into thought + a registered tool call (the e2e bench's hot
    request shape)."""
    import json as _json
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.grammar import ReActTurnGrammar
    from aurora_amd.engine.sampling_params import SamplingParams

    eng = LLMEngine(EngineConfig(model="smoke-llama", device="cuda", seed=8,
                                 kv_cache_pages=64))
    tools = ["kubectl", "final_answer", "knowledge_base_search"]
    outs = eng.generate(
        [[1, 2, 3], [4, 5, 6], [7, 8, 9]],
        SamplingParams(temperature=0.6, max_tokens=260,
                       grammar=ReActTurnGrammar(tools, thought_bytes=120,
                                                tool_soft_close=80)))
    for toks in outs:
        text = eng.tokenizer.decode(toks)
        head, sep, tail = text.rpartition("\nAction:")
        assert sep, text[-200:]
        doc = _json.loads(tail.strip())
        assert doc["tool"] in tools and isinstance(doc["args"], dict)
    assert eng.counters["grammar_fallbacks"] == 0  # byte-bucket pick covers


@pytest.mark.gpu
def test_engine_gpu_qwen_bias():
    """Qwen2-shaped model (QKV bias) through the GPU engine: generation is
    deterministic under greedy, and the bias genuinely participates in the
    forward (zeroing it changes the logits)."""
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    prompts = [[5, 6, 7, 8] * 12, [9] * 19]
    p = SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True)
    eng = LLMEngine(EngineConfig(model="smoke-qwen", device="cuda", seed=9,
                                 kv_cache_pages=64))
    outs = eng.generate(prompts, p)
    assert all(len(o) == 8 for o in outs)
    eng2 = LLMEngine(EngineConfig(model="smoke-qwen", device="cuda", seed=9,
                                  kv_cache_pages=64))
    assert eng2.generate(prompts, p) == outs
    # the bias genuinely participates: zeroing it moves the prefill logits
    from aurora_amd.models.llama import AttnMetadata
    L = 24
    ids = torch.arange(3, 3 + L, device="cuda")

    def logits(model):
        kv = [(torch.zeros_like(c[0]), torch.zeros_like(c[1]))
              for c in eng2.kv.caches]
        meta = AttnMetadata(
            mode="prefill",
            positions=torch.arange(L, dtype=torch.int32, device="cuda"),
            slot_mapping=torch.arange(L, dtype=torch.int32, device="cuda"),
            block_tables=torch.arange(1, dtype=torch.int32,
                                      device="cuda").unsqueeze(0),
            context_lens=torch.tensor([L], dtype=torch.int32, device="cuda"),
            cu_seqlens_q=torch.tensor([0, L], dtype=torch.int32,
                                      device="cuda"),
        )
        return model.compute_logits(model(ids, kv, meta)).float()

    before = logits(eng2.model)
    for layer in eng2.model.layers:
        assert layer.attn.qkv_proj.bias is not None
        layer.attn.qkv_proj.bias.zero_()
    after = logits(eng2.model)
    assert (before - after).abs().max().item() > 1e-3


def test_engine_gpu_prefix_cache_sharing():
    """Cross-request prefix cache on the GPU engine (P=256): a burst of
    requests sharing a 512-token prefix must attach pages (less prefill)
    and generate exactly what a cache-disabled engine does."""
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    common = (torch.Generator().manual_seed(6), )
    gen = common[0]
    prefix = torch.randint(1, 900, (512,), generator=gen).tolist()
    prompts = [prefix + torch.randint(1, 900, (20,), generator=gen).tolist()
               for _ in range(4)]
    p = SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)

    eng = LLMEngine(EngineConfig(model="smoke-llama", device="cuda", seed=4,
                                 kv_cache_pages=96))
    # sequential submits so registration precedes later attaches
    outs = [eng.generate([pr], p)[0] for pr in prompts]
    shared = eng.counters["prefix_cache_pages"]
    assert shared == 3 * 2, shared  # 3 later requests x 2 full 256-tok pages
    ref = LLMEngine(EngineConfig(model="smoke-llama", device="cuda", seed=4,
                                 kv_cache_pages=96, prefix_cache=False))
    ref_outs = [ref.generate([pr], p)[0] for pr in prompts]
    # request 1 never shares: both engines ran the identical computation
    # (suffix-only prefill batches differ in GEMM shape, so later requests
    # may round differently at bf16 — their exact-output equivalence is
    # guaranteed by the CPU fp32 suite, tests/test_prefix_cache.py)
    assert outs[0] == ref_outs[0]
    # prefill accounting: shared pages were never re-prefilled
    assert (eng.counters["prompt_tokens"]
            == ref.counters["prompt_tokens"] - shared * eng.kv.page_size)
    # the cached engine is deterministic under greedy
    eng2 = LLMEngine(EngineConfig(model="smoke-llama", device="cuda", seed=4,
                                  kv_cache_pages=96))
    assert [eng2.generate([pr], p)[0] for pr in prompts] == outs


def test_knn_1m_vectors():
    """BASELINE config 3 scale: kNN over a 1M-chunk HBM-resident bank
    (1.5 GB bf16) returns exact neighbours; brute-force GEMV at this
    scale is the design point (rag/index.py docstring)."""
    torch.manual_seed(2)
    N, H = 1_000_000, 768
    docs = torch.nn.functional.normalize(
        torch.randn(N, H, device="cuda", dtype=torch.float32), dim=-1
    ).to(torch.bfloat16)
    probe = [7, 499_999, 999_983]
    qv = docs[probe].float() + 1e-3 * torch.randn(len(probe), H, device="cuda")
    import time
    torch.cuda.synchronize()
    t0 = time.monotonic()
    vals, idx = ops.knn_topk(qv.to(torch.bfloat16), docs, 8)
    torch.cuda.synchronize()
    ms = (time.monotonic() - t0) * 1e3
    for row, want in enumerate(probe):
        assert idx[row, 0].item() == want, (row, idx[row, :3])
    assert ms < 2000, f"1M-vector kNN took {ms:.0f} ms"
