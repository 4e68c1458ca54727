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
