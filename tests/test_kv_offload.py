"""KV host-offload tests: full-prompt restore skips prefill and reproduces
identical greedy outputs; LRU eviction under budget."""
import pytest
import torch

from kaito_amd.engine import EngineConfig, LLMEngine, SamplingParams
from kaito_amd.models import get_model_config
from kaito_amd.parallel.state import init_parallel


@pytest.fixture(autouse=True)
def _p():
    init_parallel(1)


def _engine(**kw):
    base = dict(model=get_model_config("tiny-llama-test"), device="cpu",
                max_num_seqs=8, num_gpu_blocks=64, enforce_eager=True,
                max_model_len=128, kv_offload=True,
                kv_offload_bytes=64 << 20,
                # offload-path tests: the block-hash prefix cache would
                # intercept first (it takes priority at admission)
                enable_prefix_caching=False)
    base.update(kw)
    return LLMEngine(EngineConfig(**base))


def test_restore_skips_prefill_same_tokens():
    eng = _engine()
    prompt = list(range(20, 52))
    sp = SamplingParams(max_tokens=6, ignore_eos=True)
    first = eng.generate([prompt], sp)[0].output_token_ids
    assert eng.kv_offload.hits == 0
    # resubmit the identical prompt: restore path must hit and skip prefill
    calls = {"prefill": 0}
    orig = eng.runner.execute_prefill

    def counting(seqs):
        calls["prefill"] += 1
        return orig(seqs)

    eng.runner.execute_prefill = counting
    second = eng.generate([prompt], sp)[0].output_token_ids
    assert eng.kv_offload.hits == 1
    assert calls["prefill"] == 0, "prefill should be skipped on full hit"
    assert second == first


def test_miss_on_different_prompt():
    eng = _engine()
    sp = SamplingParams(max_tokens=4, ignore_eos=True)
    eng.generate([[1, 2, 3, 4]], sp)
    eng.generate([[1, 2, 3, 5]], sp)
    assert eng.kv_offload.hits == 0
    assert eng.kv_offload.misses >= 1


def test_lru_eviction_under_budget():
    eng = _engine(kv_offload_bytes=1 << 15)  # tiny: ~1 seq worth
    sp = SamplingParams(max_tokens=2, ignore_eos=True)
    p1 = list(range(10, 30))
    p2 = list(range(40, 60))
    eng.generate([p1], sp)
    eng.generate([p2], sp)  # evicts p1's entry
    eng.generate([p1], sp)
    assert eng.kv_offload.hits == 0  # p1 was evicted
    assert eng.kv_offload.used_bytes <= 1 << 15


def test_partial_prefix_restore_matches_fresh():
    """A prompt sharing a 32-token prefix with a cached sequence restores
    the prefix and context-prefills only the suffix — outputs must match a
    fresh engine exactly."""
    base = list(range(100, 140))            # 40-token prompt A
    sp = SamplingParams(max_tokens=5, ignore_eos=True)
    eng = _engine(seed=5)
    eng.generate([base], sp)                # A + gen now cached

    new_prompt = base[:32] + list(range(300, 316))   # A[:32] + C
    calls = {"ctx": 0}
    import kaito_amd.ops as O
    orig_ctx = O.context_attention

    def counting(*a, **k):
        calls["ctx"] += 1
        return orig_ctx(*a, **k)

    O.context_attention = counting
    try:
        out = eng.generate([new_prompt], sp)[0].output_token_ids
    finally:
        O.context_attention = orig_ctx
    assert eng.kv_offload.hits >= 1
    assert calls["ctx"] > 0, "suffix prefill should use context attention"

    fresh = _engine(seed=5)
    expect = fresh.generate([new_prompt], sp)[0].output_token_ids
    assert out == expect
