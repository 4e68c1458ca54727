"""Engine integration tests (CPU): continuous batching == naive generation."""
import pytest
import torch

from aurora_amd.engine.engine import EngineConfig, LLMEngine
from aurora_amd.engine.grammar import JsonGrammar, validate_json_bytes
from aurora_amd.engine.sampling_params import SamplingParams
from aurora_amd.models.config import get_spec
from aurora_amd.models.llama import AttnMetadata, LlamaForCausalLM


def naive_greedy(model, prompt, n_tokens):
    """Full recompute each step through a scratch cache (oracle path)."""
    spec = model.spec
    ids = list(prompt)
    P = 16
    for _ in range(n_tokens):
        L = len(ids)
        npages = (L + P - 1) // P
        kc = [torch.zeros(npages, spec.n_kv_heads, P, spec.head_dim) for _ in range(spec.n_layers)]
        vc = [torch.zeros(npages, spec.n_kv_heads, P, spec.head_dim) for _ in range(spec.n_layers)]
        caches = list(zip(kc, vc))
        meta = AttnMetadata(
            mode="prefill",
            positions=torch.arange(L, dtype=torch.int32),
            slot_mapping=torch.arange(L, dtype=torch.int32),
            block_tables=torch.arange(npages, dtype=torch.int32).unsqueeze(0),
            context_lens=torch.tensor([L], dtype=torch.int32),
            cu_seqlens_q=torch.tensor([0, L], dtype=torch.int32),
        )
        hidden = model(torch.tensor(ids), caches, meta)
        logits = model.compute_logits(hidden[-1:])
        ids.append(int(logits[0].argmax()))
    return ids[len(prompt):]


@pytest.fixture(scope="module")
def engine():
    return LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=7))


def test_greedy_matches_naive(engine):
    prompt = [5, 9, 2, 44, 17, 80, 3]
    naive = naive_greedy(engine.model, prompt, 12)
    out = engine.generate([prompt], SamplingParams(temperature=0.0, max_tokens=12, ignore_eos=True))
    assert out[0] == naive


def test_batched_matches_single(engine):
    prompts = [[5, 9, 2], list(range(40, 80)), [7] * 25]
    singly = [
        engine.generate([p], SamplingParams(temperature=0.0, max_tokens=10, ignore_eos=True))[0]
        for p in prompts
    ]
    batched = engine.generate(prompts, SamplingParams(temperature=0.0, max_tokens=10, ignore_eos=True))
    assert batched == singly


def test_chunked_prefill_matches_unchunked():
    big = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=7))
    small = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=7, max_batched_tokens=8))
    prompt = list(range(10, 60))
    p = SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True)
    assert big.generate([prompt], p) == small.generate([prompt], p)


def test_session_kv_reuse_matches_fresh():
    cfg = EngineConfig(model="tiny-llama", device="cpu", seed=7)
    e1 = LLMEngine(cfg)
    p = SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)
    turn1 = list(range(5, 37))
    e1.add_request("r1", turn1, p, session_id="sess")
    while e1.has_work():
        e1.step()
    # second turn extends the first
    turn2 = turn1 + [100, 101, 102, 103]
    e1.add_request("r2", turn2, p, session_id="sess")
    outs2 = []
    while e1.has_work():
        for o in e1.step():
            outs2.extend(o.new_token_ids)
    # fresh engine, no session reuse
    e2 = LLMEngine(cfg)
    fresh = e2.generate([turn2], p)[0]
    assert outs2 == fresh
    # and reuse really skipped recompute of whole pages
    assert e1.counters["prompt_tokens"] < len(turn1) + len(turn2)


def test_preemption_recovers():
    # tiny KV budget forces eviction/requeue; outputs must still complete
    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=7, kv_cache_pages=8))
    p = SamplingParams(temperature=0.0, max_tokens=20, ignore_eos=True)
    prompts = [list(range(2, 34)), list(range(40, 72)), list(range(80, 112))]
    outs = eng.generate(prompts, p)
    assert all(len(o) == 20 for o in outs)
    ref = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=7)).generate(prompts, p)
    assert outs == ref


def test_grammar_decoding_yields_valid_json():
    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=3))
    p = SamplingParams(temperature=0.7, max_tokens=300, grammar=JsonGrammar())
    out = eng.generate([[1, 2, 3]], p)[0]
    data = bytes(t for t in out if t < 256)
    assert validate_json_bytes(data), data


def test_stop_string():
    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=7))
    # find what greedy emits, then stop on its first two bytes
    p0 = SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)
    toks = eng.generate([[9, 8, 7]], p0)[0]
    text = bytes(t for t in toks if t < 256)
    if len(text) >= 2:
        stop = text[:2].decode("latin-1")
        p1 = SamplingParams(temperature=0.0, max_tokens=50, stop=[stop], ignore_eos=True)
        out = eng.generate([[9, 8, 7]], p1)[0]
        assert len(out) <= len(toks)


def test_abort():
    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu"))
    eng.add_request("a", [1, 2, 3], SamplingParams(max_tokens=50, ignore_eos=True))
    eng.step()
    eng.abort_request("a")
    assert not eng.has_work()
    assert eng.kv.num_free() == eng.kv.num_pages


def test_moe_model_generates():
    """Mixtral-style MoE path through the whole engine (tiny-moe, CPU)."""
    eng = LLMEngine(EngineConfig(model="tiny-moe", device="cpu", seed=5))
    out = eng.generate([[3, 1, 4, 1, 5]],
                       SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True))
    assert len(out[0]) == 6
    # deterministic
    eng2 = LLMEngine(EngineConfig(model="tiny-moe", device="cpu", seed=5))
    assert eng2.generate([[3, 1, 4, 1, 5]],
                         SamplingParams(temperature=0.0, max_tokens=6,
                                        ignore_eos=True)) == out


def test_moe_routing_uses_multiple_experts():
    import torch
    from aurora_amd.models.config import get_spec
    from aurora_amd.models.llama import MoEMLP
    spec = get_spec("tiny-moe")
    m = MoEMLP(spec, torch.float32)
    for p in m.parameters():
        torch.nn.init.normal_(p, std=0.1)
    x = torch.randn(40, spec.hidden_size)
    logits = m.router(x).float()
    sel = logits.topk(spec.experts_per_token, dim=-1).indices
    assert sel.unique().numel() > 1  # routing spreads across experts
    y = m(x)
    assert y.shape == x.shape and torch.isfinite(y).all()


def test_mixed_batch_equivalence():
    """A request arriving mid-decode (mixed prefill+decode steps) must not
    change the in-flight sequence's greedy output."""
    cfg = EngineConfig(model="tiny-llama", device="cpu", seed=7)
    p = SamplingParams(temperature=0.0, max_tokens=14, ignore_eos=True)
    a_prompt, b_prompt = list(range(5, 45)), list(range(60, 100))
    # reference: run A alone
    ref_a = LLMEngine(cfg).generate([a_prompt], p)[0]
    ref_b = LLMEngine(cfg).generate([b_prompt], p)[0]
    # mixed: start A, inject B after a few decode steps
    eng = LLMEngine(cfg)
    eng.add_request("A", a_prompt, p)
    outs = {"A": [], "B": []}
    steps = 0
    while eng.has_work():
        if steps == 4:
            eng.add_request("B", b_prompt, p)
        for o in eng.step():
            outs[o.request_id].extend(o.new_token_ids)
        steps += 1
    assert outs["A"] == ref_a
    assert outs["B"] == ref_b
    # mixed steps actually happened (prefill of B overlapped A's decode)
    assert eng.counters["prefill_steps"] >= 2



def test_moe_batched_matches_loop():
    import torch
    from aurora_amd.models.config import get_spec
    from aurora_amd.models.llama import MoEMLP
    spec = get_spec("tiny-moe")
    m = MoEMLP(spec, torch.float32)
    for p in m.parameters():
        torch.nn.init.normal_(p, std=0.1)
    x = torch.randn(33, spec.hidden_size)
    logits = m.router(x).float()
    w, sel = logits.topk(spec.experts_per_token, dim=-1)
    w = w.softmax(-1)
    batched = m._forward_batched(x, w, sel)
    # loop reference
    ref = torch.zeros_like(x)
    for e in range(spec.n_experts):
        tok, slot = (sel == e).nonzero(as_tuple=True)
        if tok.numel():
            ref.index_add_(0, tok, m.experts[e](x[tok]) * w[tok, slot, None])
    assert torch.allclose(batched, ref, atol=1e-5), (batched - ref).abs().max()


def test_engine_determinism_across_instances():
    """Same seed + same requests => identical token streams across two
    fresh engines (reproducibility; CPU)."""
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    def run():
        eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=42))
        p = SamplingParams(temperature=0.8, top_p=0.9, max_tokens=12,
                           ignore_eos=True)
        for i in range(3):
            eng.add_request(f"d{i}", [7 + i, 9, 11], p)
        toks = {f"d{i}": [] for i in range(3)}
        for _ in range(60):
            for o in eng.step():
                toks[o.request_id].extend(o.new_token_ids)
            if not eng.running and not eng.waiting:
                break
        return toks

    assert run() == run()
