"""Concurrency stress — the air-gapped analog of the reference's Go
`-race` CI tier (SURVEY.md §5 "race detection"). Hammers the shared
serving surfaces from many threads at once and asserts (a) nothing
deadlocks or raises, (b) greedy results stay DETERMINISTIC per prompt
under arbitrary interleaving, (c) aborted streams never wedge the
engine loop.
"""
import asyncio
import threading

import pytest

from kaito_amd.engine import EngineConfig, LLMEngine, SamplingParams
from kaito_amd.models import get_model_config
from kaito_amd.server.async_engine import AsyncLLMEngine


def _engine(**kw):
    d = dict(model=get_model_config("tiny-llama-test"), device="cpu",
             max_num_seqs=16, max_model_len=128, enforce_eager=True,
             num_gpu_blocks=256)
    d.update(kw)
    return LLMEngine(EngineConfig(**d))


@pytest.mark.timeout(120)
def test_async_engine_many_threads_deterministic():
    """8 submitter threads x 4 requests each (5 distinct prompts, greedy)
    through one AsyncLLMEngine: every stream completes and repeated
    prompts yield byte-identical token sequences."""
    eng = _engine()
    aeng = AsyncLLMEngine(eng).start()
    prompts = [[3 + i, 7, 11, 15] for i in range(5)]
    results = {}
    lock = threading.Lock()
    errors = []

    def worker(widx):
        async def run():
            for r in range(4):
                p = prompts[(widx + r) % len(prompts)]
                toks = []
                async for item in aeng.generate(
                        list(p), SamplingParams(max_tokens=6,
                                                ignore_eos=True)):
                    if not item.finished:
                        toks.append(item.token_id)
                with lock:
                    results.setdefault(tuple(p), []).append(tuple(toks))
        try:
            asyncio.run(run())
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=90)
        assert not t.is_alive(), "submitter thread wedged"
    aeng.shutdown()
    assert not errors, errors
    assert sum(len(v) for v in results.values()) == 32
    for p, outs in results.items():
        assert len(set(outs)) == 1, f"nondeterministic greedy for {p}: " \
                                    f"{set(outs)}"
        assert len(outs[0]) == 6


@pytest.mark.timeout(120)
def test_async_engine_churn_with_aborts():
    """Concurrent completions and mid-stream abandons (consumer walks
    away) must leave the scheduler empty and the loop alive."""
    eng = _engine()
    aeng = AsyncLLMEngine(eng).start()
    errors = []

    def worker(widx):
        async def run():
            for r in range(5):
                sp = SamplingParams(max_tokens=32, ignore_eos=True)
                n = 0
                async for item in aeng.generate([2 + widx, 9, r + 1], sp):
                    n += 1
                    if (widx + r) % 2 == 0 and n >= 2:
                        break            # abandon mid-stream
        try:
            asyncio.run(run())
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=90)
        assert not t.is_alive()
    assert not errors, errors
    # aborts drain; nothing left running
    import time
    deadline = time.monotonic() + 10
    while time.monotonic() < deadline and eng.scheduler.has_work():
        time.sleep(0.05)
    assert not eng.scheduler.has_work()
    # the loop survived: one more request completes
    async def one():
        toks = []
        async for item in aeng.generate(
                [5, 5, 5], SamplingParams(max_tokens=3, ignore_eos=True)):
            if not item.finished:
                toks.append(item.token_id)
        return toks
    assert len(asyncio.run(one())) == 3
    aeng.shutdown()


@pytest.mark.timeout(120)
def test_prefix_pool_concurrent_readers():
    """Prefix-caching pool: concurrent generate() callers sharing a hot
    prefix must all return the same greedy continuation."""
    eng = _engine(enable_prefix_caching=True, max_num_seqs=8)
    aeng = AsyncLLMEngine(eng).start()
    base = list(range(4, 36))            # 2 full blocks shared prefix
    outs = []
    lock = threading.Lock()
    errors = []

    def worker(tail):
        async def run():
            toks = []
            async for item in aeng.generate(
                    base + [tail], SamplingParams(max_tokens=4,
                                                  ignore_eos=True)):
                if not item.finished:
                    toks.append(item.token_id)
            with lock:
                outs.append((tail, tuple(toks)))
        try:
            asyncio.run(run())
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    # two waves: the second wave hits revived prefix blocks
    for wave in range(2):
        threads = [threading.Thread(target=worker, args=(100 + i,))
                   for i in range(4)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=60)
            assert not t.is_alive()
    aeng.shutdown()
    assert not errors, errors
    by_tail = {}
    for tail, toks in outs:
        by_tail.setdefault(tail, set()).add(toks)
    for tail, variants in by_tail.items():
        assert len(variants) == 1, f"tail {tail}: {variants}"
