"""LoRA multi-adapter serving tests (BASELINE config #4a class)."""
import torch

import pytest

from kaito_amd.engine import EngineConfig, LLMEngine, SamplingParams
from kaito_amd.engine.lora import LoRAAdapter, LoRAManager
from kaito_amd.models import get_model_config
from kaito_amd.models.llama import LlamaForCausalLM
from kaito_amd.parallel.state import init_parallel


@pytest.fixture(autouse=True)
def _parallel():
    init_parallel(1)


def _engine(**kw):
    base = dict(model=get_model_config("tiny-llama-test"), device="cpu",
                max_num_seqs=8, num_gpu_blocks=64, enforce_eager=True,
                max_model_len=128, enable_lora=True, max_lora_rank=16)
    base.update(kw)
    return LLMEngine(EngineConfig(**base))


def test_manager_registers_and_stacks():
    model = LlamaForCausalLM(get_model_config("tiny-llama-test")).random_init(0)
    mgr = LoRAManager(model, max_adapters=4, max_rank=16)
    s0 = mgr.register_random("a0", rank=4, seed=1)
    s1 = mgr.register_random("a1", rank=8, seed=2)
    assert (s0, s1) == (0, 1)
    assert mgr.slot("a0") == 0 and mgr.slot("missing") == -1
    key = "layers.0.self_attn.qkv_proj"
    assert mgr.stacks[key]["A"].shape[0] == 4
    # rank padding zeroed
    assert mgr.stacks[key]["A"][0, 4:].abs().sum() == 0


def test_apply_matches_manual():
    torch.manual_seed(0)
    model = LlamaForCausalLM(get_model_config("tiny-llama-test")).random_init(0)
    mgr = LoRAManager(model, max_adapters=2, max_rank=8)
    A = torch.randn(4, 256) * 0.1
    B = torch.randn(512, 4) * 0.1
    key = "layers.0.self_attn.qkv_proj"
    mgr.register(LoRAAdapter("x", 4, 8.0, {key: (A, B)}))
    xs = torch.randn(3, 256, dtype=torch.bfloat16)
    y = torch.zeros(3, 512, dtype=torch.bfloat16)
    ids = torch.tensor([0, -1, 0], dtype=torch.int32)
    out = mgr.apply(key, xs, y.clone(), ids)
    expect0 = (8.0 / 4) * (B.float() @ (A.float() @ xs[0].float()))
    assert torch.allclose(out[0].float(), expect0, atol=0.2, rtol=0.1)
    assert out[1].abs().sum() == 0  # ids=-1 untouched


def test_engine_lora_changes_output():
    eng = _engine()
    eng.runner.lora_manager.register_random("ad1", rank=8, seed=7, scale=0.2)
    prompt = [3, 14, 15, 92, 65]
    sp = SamplingParams(max_tokens=8, ignore_eos=True)
    base = eng.generate([prompt], sp)[0].output_token_ids
    eng2 = _engine()
    eng2.runner.lora_manager.register_random("ad1", rank=8, seed=7, scale=0.2)
    with_lora = eng2.generate([prompt], sp)
    sid = eng2.add_request(prompt, sp, lora_name="ad1")
    while eng2.has_unfinished():
        eng2.step()
    lora_out = eng2.seqs[sid].output_token_ids
    assert base != lora_out, "adapter should change greedy tokens"
    # base-model request in the same engine still matches plain engine
    assert with_lora[0].output_token_ids == base


def test_engine_mixed_adapter_batch():
    """Requests with different adapters decode together without crosstalk."""
    eng = _engine()
    eng.runner.lora_manager.register_random("a", rank=4, seed=1, scale=0.3)
    eng.runner.lora_manager.register_random("b", rank=4, seed=2, scale=0.3)
    prompt = [5, 6, 7, 8]
    sp = SamplingParams(max_tokens=6, ignore_eos=True)
    ia = eng.add_request(prompt, sp, lora_name="a")
    ib = eng.add_request(prompt, sp, lora_name="b")
    i0 = eng.add_request(prompt, sp)
    while eng.has_unfinished():
        eng.step()
    oa = eng.seqs[ia].output_token_ids
    ob = eng.seqs[ib].output_token_ids
    o0 = eng.seqs[i0].output_token_ids
    # solo runs must match the batched runs exactly (no crosstalk)
    for name, expect in (("a", oa), ("b", ob), (None, o0)):
        e2 = _engine()
        e2.runner.lora_manager.register_random("a", rank=4, seed=1, scale=0.3)
        e2.runner.lora_manager.register_random("b", rank=4, seed=2, scale=0.3)
        sid = e2.add_request(prompt, sp, lora_name=name)
        while e2.has_unfinished():
            e2.step()
        assert e2.seqs[sid].output_token_ids == expect, name


def test_unknown_adapter_rejected():
    eng = _engine()
    with pytest.raises(KeyError):
        eng.add_request([1, 2], SamplingParams(), lora_name="nope")
