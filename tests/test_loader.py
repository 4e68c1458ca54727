"""Checkpoint loader tests: HF-layout safetensors round trip + TP sharding."""
import pytest
import torch

from aurora_amd.models.config import get_spec
from aurora_amd.models.llama import LlamaForCausalLM
from aurora_amd.models.loader import load_llama_safetensors, save_llama_safetensors


def test_roundtrip_dense(tmp_path):
    spec = get_spec("tiny-llama")
    m1 = LlamaForCausalLM(spec, dtype=torch.float32)
    m1.init_random(11)
    save_llama_safetensors(m1, str(tmp_path))
    m2 = LlamaForCausalLM(spec, dtype=torch.float32)
    n = load_llama_safetensors(m2, str(tmp_path))
    assert n > 0
    for (na, a), (nb, b) in zip(m1.named_parameters(), m2.named_parameters()):
        assert torch.equal(a, b), na


def test_roundtrip_moe(tmp_path):
    spec = get_spec("tiny-moe")
    m1 = LlamaForCausalLM(spec, dtype=torch.float32)
    m1.init_random(12)
    save_llama_safetensors(m1, str(tmp_path))
    m2 = LlamaForCausalLM(spec, dtype=torch.float32)
    load_llama_safetensors(m2, str(tmp_path))
    for (na, a), (nb, b) in zip(m1.named_parameters(), m2.named_parameters()):
        assert torch.equal(a, b), na


def test_missing_dir():
    spec = get_spec("tiny-llama")
    m = LlamaForCausalLM(spec, dtype=torch.float32)
    with pytest.raises(FileNotFoundError):
        load_llama_safetensors(m, "/nonexistent-path-xyz")


def test_loaded_model_generates_identically(tmp_path):
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams
    spec = get_spec("tiny-llama")
    src = LlamaForCausalLM(spec, dtype=torch.float32)
    src.init_random(13)
    save_llama_safetensors(src, str(tmp_path))
    dst = LlamaForCausalLM(spec, dtype=torch.float32)
    load_llama_safetensors(dst, str(tmp_path))
    p = SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)
    e1 = LLMEngine(EngineConfig(model="tiny-llama", device="cpu"), model=src)
    e2 = LLMEngine(EngineConfig(model="tiny-llama", device="cpu"), model=dst)
    assert e1.generate([[4, 5, 6]], p) == e2.generate([[4, 5, 6]], p)


def test_corrupt_checkpoint_fails_cleanly(tmp_path):
    """Corrupted safetensors and shape mismatches must raise a clear
    error, not load garbage weights silently (models/loader.py)."""
    import pytest as _pt
    import torch
    from aurora_amd.models.config import get_spec
    from aurora_amd.models.llama import LlamaForCausalLM
    from aurora_amd.models.loader import load_llama_safetensors, save_llama_safetensors

    spec = get_spec("tiny-llama")
    m = LlamaForCausalLM(spec, dtype=torch.float32)
    m.init_random(3)
    ck = str(tmp_path / "ck")
    save_llama_safetensors(m, ck)

    # 1. truncated file
    import glob, os
    f = glob.glob(os.path.join(ck, "*.safetensors"))[0]
    data = open(f, "rb").read()
    open(f, "wb").write(data[: len(data) // 2])
    m2 = LlamaForCausalLM(spec, dtype=torch.float32)
    with _pt.raises(Exception):
        load_llama_safetensors(m2, ck)

    # 2. wrong-architecture checkpoint against a different spec
    ck2 = str(tmp_path / "ck2")
    save_llama_safetensors(m, ck2)
    other = LlamaForCausalLM(get_spec("smoke-llama"), dtype=torch.float32)
    with _pt.raises(Exception):
        load_llama_safetensors(other, ck2)


def test_roundtrip_qwen_bias(tmp_path):
    """Qwen2-family: q/k/v biases export to HF layout and fuse back."""
    from aurora_amd.engine.engine import EngineConfig, LLMEngine
    from aurora_amd.engine.sampling_params import SamplingParams
    spec = get_spec("tiny-qwen")
    m1 = LlamaForCausalLM(spec, dtype=torch.float32)
    m1.init_random(14)
    gen = torch.Generator().manual_seed(5)
    for layer in m1.layers:  # non-constant bias: catch fuse-order mistakes
        layer.attn.qkv_proj.bias.copy_(
            torch.randn(layer.attn.qkv_proj.bias.shape, generator=gen))
    save_llama_safetensors(m1, str(tmp_path))
    m2 = LlamaForCausalLM(spec, dtype=torch.float32)
    load_llama_safetensors(m2, str(tmp_path))
    for (na, a), (nb, b) in zip(m1.named_parameters(), m2.named_parameters()):
        assert torch.equal(a, b), na
    p = SamplingParams(temperature=0.0, max_tokens=6, ignore_eos=True)
    e1 = LLMEngine(EngineConfig(model="tiny-qwen", device="cpu"), model=m1)
    e2 = LLMEngine(EngineConfig(model="tiny-qwen", device="cpu"), model=m2)
    assert e1.generate([[4, 5, 6]], p) == e2.generate([[4, 5, 6]], p)


def test_qwen_bias_checkpoint_missing_bias_fails(tmp_path):
    """A biasless (Llama-layout) checkpoint must not silently load into a
    qkv_bias spec (strict mode)."""
    llama = LlamaForCausalLM(get_spec("tiny-llama"), dtype=torch.float32)
    llama.init_random(1)
    save_llama_safetensors(llama, str(tmp_path))
    qwen = LlamaForCausalLM(get_spec("tiny-qwen"), dtype=torch.float32)
    with pytest.raises(KeyError):
        load_llama_safetensors(qwen, str(tmp_path))
