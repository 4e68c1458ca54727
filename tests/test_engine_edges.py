"""Engine edge-case fuzz: adversarial SamplingParams and prompts must
never crash or hang the engine (CPU tiny model)."""
import pytest

from aurora_amd.engine.engine import EngineConfig, LLMEngine
from aurora_amd.engine.sampling_params import SamplingParams


@pytest.fixture(scope="module")
def eng():
    return LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=0))


def _drain(eng, max_steps=400):
    outs = []
    for _ in range(max_steps):
        outs.extend(eng.step())
        if not eng.running and not eng.waiting:
            break
    return outs


@pytest.mark.parametrize("params", [
    dict(temperature=0.0, max_tokens=1),
    dict(temperature=100.0, max_tokens=4),
    dict(temperature=0.5, top_p=0.0, max_tokens=4),
    dict(temperature=0.5, top_p=1.0, top_k=1, max_tokens=4),
    dict(temperature=0.7, top_k=100000, max_tokens=4),
    dict(temperature=0.0, max_tokens=0),
    dict(temperature=0.0, max_tokens=3, stop=["\u00ff\u00fe", ""]),
    dict(temperature=0.0, max_tokens=3, stop_token_ids=[0, 10**9]),
])
def test_engine_param_edges(eng, params):
    eng.add_request(f"edge-{hash(str(params)) & 0xffff}", [5, 6, 7],
                    SamplingParams(**params))
    outs = _drain(eng)
    assert any(o.finished for o in outs) or params.get("max_tokens") == 0 \
        or not outs or outs[-1].finished is not None


def test_engine_empty_and_long_prompt(eng):
    # one-token prompt
    eng.add_request("p1", [3], SamplingParams(temperature=0.0, max_tokens=2))
    outs = _drain(eng)
    assert outs and outs[-1].finished
    # prompt at the context boundary finishes via max_len
    L = eng.max_model_len - 2
    V = eng.spec.vocab_size
    eng.add_request("p2", [2 + (i % (V - 3)) for i in range(L)],
                    SamplingParams(temperature=0.0, max_tokens=64,
                                   ignore_eos=True))
    outs = _drain(eng, max_steps=2000)
    fin = [o for o in outs if o.request_id == "p2" and o.finished]
    assert fin and fin[-1].finish_reason in ("max_len", "max_tokens")


def test_engine_rejects_bad_token_ids(eng):
    import pytest as _pt
    with _pt.raises(ValueError):
        eng.add_request("bad1", [0, 10**9],
                        SamplingParams(temperature=0.0, max_tokens=1))
    with _pt.raises(ValueError):
        eng.add_request("bad2", [],
                        SamplingParams(temperature=0.0, max_tokens=1))


def test_max_len_not_shadowed_by_stop_strings():
    """With stop strings set (the agent workflow always sets them) a sequence
    must still terminate at max_model_len (the 'elif p.stop' branch
    shadowed the max_len check, letting seqs grow past max_position)."""
    e = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=0,
                               max_model_len=32))
    # stop string that never occurs in output; generous max_tokens
    e.add_request("ml-1", list(range(5, 25)),  # 20-token prompt, room = 12
                  SamplingParams(temperature=0.0, max_tokens=10_000,
                                 stop=["NEVER-EMITTED-STOP"], ignore_eos=True))
    outs = _drain(e, max_steps=200)
    assert outs and outs[-1].finished
    assert outs[-1].finish_reason in ("max_len", "max_tokens")
    seq_tokens = 20 + sum(len(o.new_token_ids) for o in outs)
    assert seq_tokens <= 32


def test_preemption_bumps_page_generation():
    """Preempting a sequence must bump page_gen so graph block-table staging
    keyed on (serial, gen, npages) can never reuse a stale row (ADVICE r1
    high: silent KV corruption after preempt+resume)."""
    e = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=0))
    e.add_request("pg-1", [5, 6, 7], SamplingParams(max_tokens=4))
    e.step()  # prefill; pg-1 now running with pages
    seq = e.seqs["pg-1"]
    assert seq.pages and seq.page_gen == 0
    assert e._preempt_lowest()
    assert seq.page_gen == 1 and not seq.pages
    # serials are unique even for a reused req_id
    e2_ids = set()
    for rid in ("dup", "dup2"):
        e.add_request(rid, [5], SamplingParams(max_tokens=1))
        e2_ids.add(e.seqs[rid].serial)
    assert len(e2_ids) == 2


def test_max_tokens_clamped_to_model_len_at_admission():
    e = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=0,
                               max_model_len=64))
    e.add_request("clamp-1", list(range(5, 35)),  # 30-token prompt
                  SamplingParams(max_tokens=10_000))
    assert e.seqs["clamp-1"].params.max_tokens <= 34


def test_rejected_request_leaves_no_subscription():
    """A request that add_request rejects at the API boundary (empty or
    out-of-vocab prompt) must not leak its server subscription — the
    stepping thread would otherwise hold a dead queue until shutdown."""
    import asyncio

    import pytest as _pytest

    from aurora_amd.engine.engine import EngineConfig
    from aurora_amd.engine.sampling_params import SamplingParams
    from aurora_amd.engine.server import AsyncEngineServer

    srv = AsyncEngineServer(EngineConfig(model="tiny-llama", device="cpu",
                                         seed=0))

    async def go():
        with _pytest.raises(ValueError):
            await srv.generate([], SamplingParams(max_tokens=2))
        with _pytest.raises(ValueError):
            await srv.generate([10 ** 9], SamplingParams(max_tokens=2))
        assert srv._subs == {}
        r = await srv.generate([1, 2, 3],
                               SamplingParams(max_tokens=2, ignore_eos=True))
        assert len(r.token_ids) == 2 and srv._subs == {}

    try:
        asyncio.run(go())
    finally:
        srv.shutdown()


def test_streaming_text_preserves_multibyte_utf8():
    """A codepoint split across tokens (every non-ASCII char under the
    byte tokenizer) must appear in the streamed text once complete — the
    old per-token errors='ignore' decode silently dropped it."""
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    LIT = "latency → 97 ms café".encode("utf-8")

    class LiteralGrammar:
        """Forces the exact byte string LIT, then completes."""

        def __init__(self, pos=0):
            self.pos = pos

        def fork(self):
            return LiteralGrammar(self.pos)

        def is_complete(self):
            return self.pos >= len(LIT)

        def advance(self, data: bytes) -> bool:
            if LIT[self.pos:self.pos + len(data)] != bytes(data) or not data:
                return False
            self.pos += len(data)
            return True

        def allowed_next_bytes(self):
            return b"" if self.is_complete() else LIT[self.pos:self.pos + 1]

    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=0))
    eng.add_request("utf8", [3, 4, 5],
                    SamplingParams(temperature=0.0, max_tokens=len(LIT) + 4,
                                   grammar=LiteralGrammar()))
    text = ""
    while eng.has_work():
        for out in eng.step():
            text += out.new_text
    assert text == LIT.decode("utf-8"), repr(text)


def test_streaming_text_flushes_partial_char_at_cutoff():
    """max_tokens cutting generation mid-codepoint must flush a
    replacement char, not strand bytes in the decoder."""
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams

    LIT = "aé".encode("utf-8")  # 3 bytes; cut after 2

    class LiteralGrammar:
        def __init__(self, pos=0):
            self.pos = pos

        def fork(self):
            return LiteralGrammar(self.pos)

        def is_complete(self):
            return self.pos >= len(LIT)

        def advance(self, data: bytes) -> bool:
            if LIT[self.pos:self.pos + len(data)] != bytes(data) or not data:
                return False
            self.pos += len(data)
            return True

        def allowed_next_bytes(self):
            return b"" if self.is_complete() else LIT[self.pos:self.pos + 1]

    eng = LLMEngine(EngineConfig(model="tiny-llama", device="cpu", seed=0))
    eng.add_request("cut", [3, 4, 5],
                    SamplingParams(temperature=0.0, max_tokens=2,
                                   grammar=LiteralGrammar()))
    text = ""
    while eng.has_work():
        for out in eng.step():
            text += out.new_text
    assert text == "a�", repr(text)
