"""Property-based invariants for the paged-KV pool + prefix cache
(hypothesis): under arbitrary interleavings of allocate / free / fork /
match_prefix / register_prefix, no block is ever in two live owners, the
free+cached+live accounting always conserves the pool, and every cache
hit serves exactly the registered content."""
import hypothesis.strategies as st
from hypothesis import given, settings

from kaito_amd.engine.block_pool import BlockPool, PrefixCachingPool

BS = 16


def _invariants(pool: PrefixCachingPool):
    live = set(pool._refcount)
    cached = set(pool._cached)
    free = set(pool._free)
    assert not live & cached, "block both live and parked"
    assert not live & free, "block both live and free"
    assert not cached & free, "block both parked and free"
    assert len(live) + len(cached) + len(free) == pool.num_blocks
    for blk, h in pool._cached.items():
        # a parked block is revivable only while the table points at it
        assert blk in pool._hash_of
    assert all(rc > 0 for rc in pool._refcount.values())


@settings(max_examples=200, deadline=None)
@given(st.lists(st.tuples(st.integers(0, 3), st.integers(0, 6),
                          st.integers(0, 5)), min_size=1, max_size=60))
def test_prefix_pool_random_interleavings(ops):
    pool = PrefixCachingPool(24, BS)
    owners = {}           # owner id → (blocks, token_ids)
    prompts = [tuple(range(s, s + BS * (1 + s % 3))) for s in range(6)]
    next_id = 0
    for op, arg, arg2 in ops:
        if op == 0:      # admit: match prefix then allocate the rest
            toks = prompts[arg % len(prompts)]
            need = (len(toks) + BS) // BS + 1
            shared, covered = pool.match_prefix(list(toks))
            rest = need - len(shared)
            if pool.can_allocate(rest):
                blocks = shared + pool.allocate(rest)
                owners[next_id] = (blocks, toks)
                next_id += 1
            else:
                pool.free(shared)      # admission failed: return shares
        elif op == 1 and owners:      # finish + register prefix
            oid = sorted(owners)[arg % len(owners)]
            blocks, toks = owners.pop(oid)
            nfull = len(toks) // BS
            pool.register_prefix(list(toks), blocks[:nfull])
            pool.free(blocks)
        elif op == 2 and owners:      # abort (no registration)
            oid = sorted(owners)[arg % len(owners)]
            blocks, _ = owners.pop(oid)
            pool.free(blocks)
        elif op == 3 and owners:      # fork a shared block (refcount)
            oid = sorted(owners)[arg % len(owners)]
            blocks, toks = owners[oid]
            blk = blocks[arg2 % len(blocks)]
            pool.fork(blk)
            pool.free([blk])          # immediately undo: rc round-trip
        _invariants(pool)
    # drain everything: accounting must return to a full pool
    for oid in list(owners):
        blocks, _ = owners.pop(oid)
        pool.free(blocks)
    _invariants(pool)
    while pool._cached:
        pool.allocate(1)              # evict-by-allocate must always work
    assert pool.num_free + len(pool._refcount) == pool.num_blocks


@settings(max_examples=100, deadline=None)
@given(st.integers(1, 6), st.integers(1, 6))
def test_plain_pool_conservation(a, b):
    p = BlockPool(12, BS)
    x = p.allocate(a)
    y = p.allocate(min(b, p.num_free))
    for blk in x:
        p.fork(blk)
    p.free(x)
    p.free(x)                         # second free clears the fork
    p.free(y)
    assert p.num_free == 12 and not p._refcount


# ---- quantization properties (same file: fast property suite) ----------
import numpy as np  # noqa: E402
import torch  # noqa: E402
from hypothesis import given as _given  # noqa: E402


@settings(max_examples=50, deadline=None)
@_given(st.integers(1, 8), st.integers(1, 4), st.integers(0, 2**31 - 1),
        st.floats(0.01, 100.0))
def test_w4_quantization_error_bounded(nmul, kmul, seed, scale):
    """For any weight distribution/scale, 4-bit group dequant error stays
    within half a quantization step everywhere."""
    from kaito_amd.models.quant import quantize_w4
    from kaito_amd.ops import torch_ref
    g = torch.Generator().manual_seed(seed)
    N, K, G = 8 * nmul, 64 * kmul, 64
    w = torch.randn(N, K, generator=g) * scale
    qw, s, z = quantize_w4(w, G)
    deq = torch_ref.w4a16_unpack(qw, s, z, G)
    step = s.repeat_interleave(G, dim=1)
    assert ((deq - w).abs() <= step * 0.5 + 1e-5 * scale).all()


@settings(max_examples=50, deadline=None)
@_given(st.integers(0, 2**31 - 1), st.integers(1, 20), st.floats(0.2, 2.0))
def test_sampler_topk_topp_support(seed, k, temp):
    """With top_k set, the sampled token must come from the top-k logits;
    with top_p, from the minimal nucleus prefix (plus the argmax row when
    tied). Greedy rows always return the argmax."""
    from kaito_amd.engine.sampler import Sampler
    from kaito_amd.engine.sequence import Sequence, SamplingParams
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(3, 64, generator=g)
    seqs = [Sequence(0, [1], SamplingParams(temperature=temp, top_k=k,
                                            seed=seed)),
            Sequence(1, [1], SamplingParams(temperature=0.0)),
            Sequence(2, [1], SamplingParams(temperature=temp, top_p=0.5,
                                            seed=seed))]
    s = Sampler("cpu")
    toks = s.sample(logits.clone(), seqs)
    topk_ids = set(torch.topk(logits[0], min(k, 64)).indices.tolist())
    assert int(toks[0]) in topk_ids
    assert int(toks[1]) == int(logits[1].argmax())
    # nucleus: token must be inside the smallest prefix with mass >= 0.5
    probs = torch.softmax(logits[2] / temp, dim=-1)
    sp, si = torch.sort(probs, descending=True)
    cum = torch.cumsum(sp, 0)
    ncut = int((cum < 0.5).sum()) + 1
    assert int(toks[2]) in set(si[:ncut].tolist())


@settings(max_examples=10, deadline=None)
@_given(st.integers(0, 2**31 - 1), st.integers(8, 40),
        st.sampled_from([True, False]))
def test_engine_oracle_under_random_workloads(seed, budget, mixed):
    """Capstone property: for random prompt sets, chunk budgets and both
    stepping modes, every sequence's greedy output equals the standalone
    full-recompute oracle."""
    import numpy as np
    from kaito_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from kaito_amd.models import get_model_config
    from tests.test_engine import _naive_generate
    rng = np.random.default_rng(seed)
    cfg = EngineConfig(model=get_model_config("tiny-llama-test"),
                       device="cpu", max_num_seqs=4, num_gpu_blocks=64,
                       enforce_eager=True, max_model_len=128,
                       max_num_batched_tokens=budget,
                       enable_mixed_batch=mixed, mixed_prefill_tokens=16)
    eng = LLMEngine(cfg)
    prompts = [rng.integers(1, 500, rng.integers(3, 60)).tolist()
               for _ in range(int(rng.integers(1, 5)))]
    sp = SamplingParams(max_tokens=int(rng.integers(2, 8)),
                        ignore_eos=True)
    outs = eng.generate(prompts, sp)
    for p, o in zip(prompts, outs):
        assert o.output_token_ids == _naive_generate(
            eng.runner.model, cfg, p, sp.max_tokens)
