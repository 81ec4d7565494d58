"""MLA (DeepSeek multi-head latent attention) unit tests — CPU.

The absorbed decode identity is the load-bearing property: scores over
the latent cache (q_c·c_kv + q_pe·k_pe) must equal the non-absorbed
per-head attention (q_nope·k_nope + q_pe·k_pe), and the latent-space V
accumulation followed by W_uv must equal attention over the decompressed
per-head V. GPU kernel numerics live in tests/test_gpu_kernels.py.
"""
import pytest
import torch

from kaito_amd.engine import EngineConfig, LLMEngine, SamplingParams
from kaito_amd.engine.config import ModelConfig
from kaito_amd.models import get_model_config
from kaito_amd.ops import torch_ref


def test_mla_decode_ref_matches_dense_attention():
    """torch_ref.mla_decode == dense softmax attention in latent space."""
    torch.manual_seed(0)
    T, H, R, P, BS = 3, 4, 32, 16, 16
    DT = R + P
    L = [20, 33, 7]
    nb = [(x + BS - 1) // BS for x in L]
    NB = sum(nb) + 1
    cache = torch.randn(NB, BS, DT)
    bt = torch.zeros(T, max(nb), dtype=torch.int32)
    nxt = 1
    for i in range(T):
        bt[i, :nb[i]] = torch.arange(nxt, nxt + nb[i], dtype=torch.int32)
        nxt += nb[i]
    sl = torch.tensor(L, dtype=torch.int32)
    q = torch.randn(T, H, DT)
    scale = 0.17
    out = torch_ref.mla_decode(q, cache, bt, sl, scale, R)
    for i in range(T):
        rows = cache[bt[i, :nb[i]].long()].reshape(-1, DT)[:L[i]]
        p = torch.softmax(q[i] @ rows.t() * scale, dim=-1)
        torch.testing.assert_close(out[i], p @ rows[:, :R],
                                   rtol=1e-4, atol=1e-4)


def test_mla_absorbed_equals_nonabsorbed():
    """Absorption is exact linear algebra: folding W_uk into q and
    applying W_uv after the latent accumulation equals the standard
    per-head attention over decompressed K/V."""
    torch.manual_seed(1)
    T, H = 5, 2
    R, NOPE, PE, V = 24, 12, 8, 10
    c_kv = torch.randn(T, R)            # cached latents (one seq)
    k_pe = torch.randn(T, PE)
    w_kc = torch.randn(H, NOPE, R) * 0.3
    w_vc = torch.randn(H, R, V) * 0.3
    q_nope = torch.randn(1, H, NOPE)    # one decode token
    q_pe = torch.randn(1, H, PE)
    scale = (NOPE + PE) ** -0.5

    # non-absorbed: decompress K/V per head
    k_nope = torch.einsum("hnr,tr->thn", w_kc, c_kv)
    v = torch.einsum("hrv,tr->thv", w_vc, c_kv)
    s = (torch.einsum("bhn,thn->bht", q_nope, k_nope)
         + torch.einsum("bhp,tp->bht", q_pe, k_pe.unsqueeze(0)[0])) * scale
    p = torch.softmax(s, dim=-1)
    want = torch.einsum("bht,thv->bhv", p, v)

    # absorbed: latent-space scores and accumulation
    q_c = torch.einsum("bhn,hnr->bhr", q_nope, w_kc)
    s2 = (torch.einsum("bhr,tr->bht", q_c, c_kv)
          + torch.einsum("bhp,tp->bht", q_pe, k_pe)) * scale
    p2 = torch.softmax(s2, dim=-1)
    out_c = torch.einsum("bht,tr->bhr", p2, c_kv)
    got = torch.einsum("bhr,hrv->bhv", out_c, w_vc)
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-5)


def _cfg(**kw):
    mc = get_model_config("tiny-deepseek-test")
    d = dict(model=mc, device="cpu", max_num_seqs=4, max_model_len=96,
             enforce_eager=True, num_gpu_blocks=128)
    d.update(kw)
    return EngineConfig(**d)


def test_mla_engine_config_guards():
    cfg = _cfg(enable_prefix_caching=True, kv_offload=True)
    eng = LLMEngine(cfg)
    # host offload stays off (aliased cache pair would double the bytes);
    # prefix caching + chunked prefill now work over the latent cache
    assert eng.kv_offload is None
    assert cfg.enable_prefix_caching
    # latent cache: aliased (c, c) pair, [NB+1, BS, r+rope]
    k, v = eng.runner.kv_caches[0]
    assert k.data_ptr() == v.data_ptr()
    assert k.shape[-1] == cfg.model.kv_cache_row
    assert k.dim() == 3


def test_mla_chunked_prefill_matches_whole():
    """A prompt longer than the step budget is CHUNKED (latent-cache
    context attention, MLAAttention._context) and must produce the same
    greedy tokens as an engine with a budget that fits it whole."""
    cfg = _cfg(max_num_batched_tokens=32, max_model_len=96)
    eng = LLMEngine(cfg)
    prompt = list(range(2, 80))                 # 78 tokens -> 3 chunks
    sp = SamplingParams(max_tokens=6, ignore_eos=True)
    got = eng.generate([prompt], sp)[0].output_token_ids
    cfg2 = _cfg(max_num_batched_tokens=512, max_model_len=96)
    eng2 = LLMEngine(cfg2)
    want = eng2.generate([prompt], sp)[0].output_token_ids
    assert got == want


def test_mla_prefix_cache_hit_matches_cold():
    """Prefix-cache revival over the latent cache: a repeated prompt
    skips prefill (covered blocks) and still decodes identically."""
    cfg = _cfg(enable_prefix_caching=True)
    eng = LLMEngine(cfg)
    prompt = list(range(3, 3 + 48))             # 3 full blocks
    sp = SamplingParams(max_tokens=5, ignore_eos=True)
    first = eng.generate([prompt], sp)[0].output_token_ids
    again = eng.generate([prompt], sp)[0].output_token_ids
    assert first == again


def test_mla_kv_bytes_per_token():
    mc = get_model_config("deepseek-v2-lite")
    # 27 layers * (512+64) latent * bf16, replicated across TP
    assert mc.kv_bytes_per_token(1) == 27 * 576 * 2
    assert mc.kv_bytes_per_token(8) == 27 * 576 * 2
    assert mc.is_mla and mc.rotary_dim == 64


def test_yarn_cos_sin_cache_shape_and_scale():
    from kaito_amd.models.llama import build_cos_sin_cache
    mc = get_model_config("tiny-deepseek-test")
    cs = build_cos_sin_cache(mc, "cpu", max_pos=256)
    assert cs.shape == (256, 64)        # qk_rope_head_dim
    assert torch.isfinite(cs).all()
    # yarn mscale with mscale == mscale_all_dim cancels on cos/sin
    assert abs(cs[0, 0].item() - 1.0) < 1e-5


def test_mla_preemption_recompute_coherent():
    """Preempted MLA sequences recompute prompt+generated whole (no
    context chunks) and continue with identical greedy tokens."""
    cfg = _cfg(num_gpu_blocks=20, max_num_seqs=3)
    eng = LLMEngine(cfg)
    sp = SamplingParams(max_tokens=10, ignore_eos=True)
    prompts = [[7 + i, 9, 11, 13, 15, 17, 19, 21] for i in range(3)]
    outs = eng.generate(prompts, sp)
    # oracle: fresh engine with plenty of blocks (no preemption)
    cfg2 = _cfg(num_gpu_blocks=128, max_num_seqs=3)
    eng2 = LLMEngine(cfg2)
    outs2 = eng2.generate(prompts, sp)
    for a, b in zip(outs, outs2):
        assert a.output_token_ids == b.output_token_ids


def test_deepseek_checkpoint_roundtrip(tmp_path):
    """Export a random-init tiny-deepseek engine's weights into an
    HF-DeepSeek-named safetensors checkpoint (re-INTERLEAVING the rope
    rows the way real checkpoints store them, merging w_kc/w_vc back
    into kv_b_proj), then load into a fresh engine: greedy decode must
    match exactly. Validates the loader mapping bidirectionally,
    including the rope de-interleave (models/loader.py load_mla_attn)."""
    from safetensors.torch import save_file
    mc = get_model_config("tiny-deepseek-test")
    base = LLMEngine(_cfg())
    model = base.runner.model
    params = dict(model.named_parameters())
    H, NOPE, PE, R_ = (mc.num_heads, mc.qk_nope_head_dim,
                       mc.qk_rope_head_dim, mc.kv_lora_rank)

    def interleave_pe(t, n_heads=1):
        # inverse of the loader's deinterleave: half-split -> interleaved
        d = t.shape[0] // n_heads
        x = t.reshape(n_heads, 2, d // 2, -1)
        return x.transpose(1, 2).reshape(t.shape[0], -1)

    tensors = {
        "model.embed_tokens.weight": params["embed_tokens.weight"].data,
        "model.norm.weight": params["norm"].data,
        "lm_head.weight": params["lm_head.weight"].data,
    }
    for i in range(mc.num_layers):
        pre = f"layers.{i}."
        a = pre + "self_attn."
        o = "model." + a
        tensors[o + "q_a_proj.weight"] = params[a + "q_a_proj"].data
        tensors[o + "q_a_layernorm.weight"] = params[a + "q_a_layernorm"].data
        qb = params[a + "q_b_proj.weight"].data.reshape(H, NOPE + PE, -1)
        qb_il = torch.cat(
            [qb[:, :NOPE],
             interleave_pe(qb[:, NOPE:].reshape(H * PE, -1), H
                           ).reshape(H, PE, -1)], 1)
        tensors[o + "q_b_proj.weight"] = qb_il.reshape(H * (NOPE + PE), -1)
        kva = params[a + "kv_a_proj_with_mqa"].data
        tensors[o + "kv_a_proj_with_mqa.weight"] = torch.cat(
            [kva[:R_], interleave_pe(kva[R_:])], 0)
        tensors[o + "kv_a_layernorm.weight"] = params[a + "kv_a_layernorm"].data
        kvb = torch.cat([params[a + "w_kc"].data,
                         params[a + "w_vc"].data.transpose(1, 2)], 1)
        tensors[o + "kv_b_proj.weight"] = kvb.reshape(-1, R_)
        tensors[o + "o_proj.weight"] = params[a + "o_proj.weight"].data
        for ln in ("input_layernorm", "post_attention_layernorm"):
            tensors[f"model.{pre}{ln}.weight"] = params[pre + ln].data
        if i < mc.first_k_dense:
            gu = params[pre + "mlp.gate_up_proj.weight"].data
            ii = mc.intermediate_size
            tensors[f"model.{pre}mlp.gate_proj.weight"] = gu[:ii]
            tensors[f"model.{pre}mlp.up_proj.weight"] = gu[ii:]
            tensors[f"model.{pre}mlp.down_proj.weight"] = \
                params[pre + "mlp.down_proj.weight"].data
        else:
            tensors[f"model.{pre}mlp.gate.weight"] = params[pre + "mlp.gate"].data
            tensors[f"model.{pre}mlp.gate.e_score_correction_bias"] = \
                params[pre + "mlp.e_score_correction_bias"].data
            for le in range(mc.num_experts):
                ex = f"model.{pre}mlp.experts.{le}."
                w = params[pre + "mlp.w_gate_up"].data[le]
                ie = mc.moe_intermediate_size
                tensors[ex + "gate_proj.weight"] = w[:ie]
                tensors[ex + "up_proj.weight"] = w[ie:]
                tensors[ex + "down_proj.weight"] = \
                    params[pre + "mlp.w_down"].data[le]
            sh = f"model.{pre}mlp.shared_experts."
            sgu = params[pre + "mlp.w_shared_gate_up"].data
            sie = mc.n_shared_experts * mc.moe_intermediate_size
            tensors[sh + "gate_proj.weight"] = sgu[:sie]
            tensors[sh + "up_proj.weight"] = sgu[sie:]
            tensors[sh + "down_proj.weight"] = \
                params[pre + "mlp.w_shared_down"].data

    d = tmp_path / "ckpt"
    d.mkdir()
    save_file({k: v.contiguous().clone() for k, v in tensors.items()},
              str(d / "model.safetensors"))

    prompt = [5, 9, 13, 17, 21]
    sp = SamplingParams(max_tokens=8, ignore_eos=True)
    want = base.generate([prompt], sp)[0].output_token_ids
    eng = LLMEngine(_cfg(), weights_path=str(d))
    got = eng.generate([prompt], sp)[0].output_token_ids
    assert got == want
