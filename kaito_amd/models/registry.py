"""Model catalog: preset name → ModelConfig.

MI355X-native analog of the reference's preset catalog
(/root/reference/presets/workspace/models/supported_models.yaml — 31 curated
presets; model_catalog.yaml — hiddenSize/numLayers/kvHeads metadata used by
the estimator's BytesPerToken). Architectures here are public model specs.
"""
from __future__ import annotations

from typing import Dict

from ..engine.config import ModelConfig

_REGISTRY: Dict[str, ModelConfig] = {}


def register(cfg: ModelConfig) -> ModelConfig:
    _REGISTRY[cfg.name] = cfg
    return cfg


def get_model_config(name: str, weights_path: str = None) -> ModelConfig:
    """Resolve a preset by name; unknown names fall back to dynamic
    resolution from the weights directory's config.json (the reference's
    generateHuggingFaceModel path, vllm_model.go:153 — best-effort models
    outside the curated list)."""
    key = name.lower()
    if key in _REGISTRY:
        return _REGISTRY[key]
    if weights_path:
        import json
        import os
        cfg_path = os.path.join(weights_path, "config.json")
        if os.path.exists(cfg_path):
            with open(cfg_path) as f:
                cfg = ModelConfig.from_hf_config(json.load(f), name=name)
            return register(cfg)
    raise KeyError(f"unknown model preset: {name}; known: {sorted(_REGISTRY)}")


def list_models():
    return sorted(_REGISTRY)


# ---- Llama family (BASELINE configs #2, #3) --------------------------------
register(ModelConfig(
    name="llama-3-8b", hidden_size=4096, num_layers=32, num_heads=32,
    num_kv_heads=8, intermediate_size=14336, vocab_size=128256, head_dim=128,
    rope_theta=500000.0, max_position=8192))
register(ModelConfig(
    name="llama-3.1-8b", hidden_size=4096, num_layers=32, num_heads=32,
    num_kv_heads=8, intermediate_size=14336, vocab_size=128256, head_dim=128,
    rope_theta=500000.0, max_position=131072))
register(ModelConfig(
    name="llama-3-70b", hidden_size=8192, num_layers=80, num_heads=64,
    num_kv_heads=8, intermediate_size=28672, vocab_size=128256, head_dim=128,
    rope_theta=500000.0, max_position=8192))
register(ModelConfig(
    name="llama-3.3-70b", hidden_size=8192, num_layers=80, num_heads=64,
    num_kv_heads=8, intermediate_size=28672, vocab_size=128256, head_dim=128,
    rope_theta=500000.0, max_position=131072))

# ---- Phi family (reference benchmark model: Phi-4-mini-instruct) ----------
register(ModelConfig(
    name="phi-4-mini-instruct", hidden_size=3072, num_layers=32, num_heads=24,
    num_kv_heads=8, intermediate_size=8192, vocab_size=200064, head_dim=128,
    rope_theta=10000.0, max_position=131072, partial_rotary_factor=0.75,
    tie_word_embeddings=True))
register(ModelConfig(
    name="phi-3-mini-4k-instruct", hidden_size=3072, num_layers=32,
    num_heads=32, num_kv_heads=32, intermediate_size=8192, vocab_size=32064,
    head_dim=96, rope_theta=10000.0, max_position=4096))

# ---- Mistral / Qwen dense --------------------------------------------------
register(ModelConfig(
    name="mistral-7b", hidden_size=4096, num_layers=32, num_heads=32,
    num_kv_heads=8, intermediate_size=14336, vocab_size=32000, head_dim=128,
    rope_theta=10000.0, max_position=32768))
register(ModelConfig(
    name="qwen2.5-7b", hidden_size=3584, num_layers=28, num_heads=28,
    num_kv_heads=4, intermediate_size=18944, vocab_size=152064, head_dim=128,
    rope_theta=1000000.0, max_position=32768, attention_bias=True))
register(ModelConfig(
    name="qwen2.5-72b", hidden_size=8192, num_layers=80, num_heads=64,
    num_kv_heads=8, intermediate_size=29568, vocab_size=152064, head_dim=128,
    rope_theta=1000000.0, max_position=32768, attention_bias=True))

# ---- MoE family (mixtral-class; fused-MoE presets in the reference
# catalog, e.g. supported_models.yaml mixtral/gpt-oss rows) ------------------
register(ModelConfig(
    name="mixtral-8x7b", hidden_size=4096, num_layers=32, num_heads=32,
    num_kv_heads=8, intermediate_size=14336, vocab_size=32000, head_dim=128,
    rope_theta=1000000.0, max_position=32768, num_experts=8,
    num_experts_per_tok=2, moe_intermediate_size=14336))

# ---- instruct aliases (reference preset names, supported_models.yaml) ------
def _alias(new: str, base: str, **over) -> None:
    import dataclasses
    register(dataclasses.replace(_REGISTRY[base], name=new, **over))


_alias("llama-3.1-8b-instruct", "llama-3.1-8b")
_alias("llama-3.3-70b-instruct", "llama-3.3-70b")
_alias("mistral-7b-instruct", "mistral-7b")
_alias("deepseek-r1-distill-llama-8b", "llama-3.1-8b")

# ---- remaining reference catalog (31 curated presets). Architectures are
# public model specs; presets whose architecture falls outside the HIP
# engine's llama-family support matrix (parallel-block falcon, gemma-3
# norm/sliding-window, deepseek MLA, gpt-oss attention sinks) carry
# runtime="transformers" — the same native/fallback runtime split the
# reference makes between its vLLM and text-generation runtimes. ----------

# Phi family (remaining)
register(ModelConfig(
    name="phi-3-mini-128k-instruct", hidden_size=3072, num_layers=32,
    num_heads=32, num_kv_heads=32, intermediate_size=8192, vocab_size=32064,
    head_dim=96, rope_theta=10000.0, max_position=131072))
register(ModelConfig(
    name="phi-3-medium-4k-instruct", hidden_size=5120, num_layers=40,
    num_heads=40, num_kv_heads=10, intermediate_size=17920, vocab_size=32064,
    head_dim=128, rope_theta=10000.0, max_position=4096))
register(ModelConfig(
    name="phi-3-medium-128k-instruct", hidden_size=5120, num_layers=40,
    num_heads=40, num_kv_heads=10, intermediate_size=17920, vocab_size=32064,
    head_dim=128, rope_theta=10000.0, max_position=131072))
register(ModelConfig(
    name="phi-3.5-mini-instruct", hidden_size=3072, num_layers=32,
    num_heads=32, num_kv_heads=32, intermediate_size=8192, vocab_size=32064,
    head_dim=96, rope_theta=10000.0, max_position=131072))
register(ModelConfig(
    name="phi-4", hidden_size=5120, num_layers=40, num_heads=40,
    num_kv_heads=10, intermediate_size=17920, vocab_size=100352,
    head_dim=128, rope_theta=250000.0, max_position=16384))
# phi-2: parallel attn+MLP block, LayerNorm, ungated GELU MLP, partial
# rotary. head_dim 80 is PADDED to 128 for the D∈{64,128,256} attention
# kernels (zero-padded dims contribute nothing; attn_scale keeps 80^-0.5;
# rotary covers the original 32 dims → 32/128 = 0.25).
register(ModelConfig(
    name="phi-2", hidden_size=2560, num_layers=32, num_heads=32,
    num_kv_heads=32, intermediate_size=10240, vocab_size=51200, head_dim=128,
    rope_theta=10000.0, max_position=2048, partial_rotary_factor=0.25,
    attn_scale=1.0 / 80 ** 0.5, parallel_block=True, norm_type="layernorm",
    gated_mlp=False, hidden_act="gelu", attention_bias=True))

# Mistral family (remaining); ministral-3 dims from published model cards
register(ModelConfig(
    name="ministral-3-3b-instruct", hidden_size=3072, num_layers=26,
    num_heads=24, num_kv_heads=8, intermediate_size=8192, vocab_size=131072,
    head_dim=128, rope_theta=1000000.0, max_position=131072))
register(ModelConfig(
    name="ministral-3-8b-instruct", hidden_size=4096, num_layers=34,
    num_heads=32, num_kv_heads=8, intermediate_size=12288, vocab_size=131072,
    head_dim=128, rope_theta=1000000.0, max_position=131072))
register(ModelConfig(
    name="ministral-3-14b-instruct", hidden_size=5120, num_layers=40,
    num_heads=40, num_kv_heads=8, intermediate_size=16384, vocab_size=131072,
    head_dim=128, rope_theta=1000000.0, max_position=131072))
register(ModelConfig(
    name="mistral-large-3-675b-instruct", hidden_size=7168, num_layers=61,
    num_heads=128, num_kv_heads=128, intermediate_size=18432,
    vocab_size=131072, head_dim=64, rope_theta=1000000.0,
    max_position=131072, num_experts=256, num_experts_per_tok=8,
    moe_intermediate_size=2048, runtime="transformers"))  # MLA-class MoE

# Qwen family (remaining)
register(ModelConfig(
    name="qwen2.5-coder-7b-instruct", hidden_size=3584, num_layers=28,
    num_heads=28, num_kv_heads=4, intermediate_size=18944, vocab_size=152064,
    head_dim=128, rope_theta=1000000.0, max_position=32768,
    attention_bias=True))
register(ModelConfig(
    name="qwen2.5-coder-32b-instruct", hidden_size=5120, num_layers=64,
    num_heads=40, num_kv_heads=8, intermediate_size=27648, vocab_size=152064,
    head_dim=128, rope_theta=1000000.0, max_position=32768,
    attention_bias=True))
register(ModelConfig(
    name="deepseek-r1-distill-qwen-14b", hidden_size=5120, num_layers=48,
    num_heads=40, num_kv_heads=8, intermediate_size=13824, vocab_size=152064,
    head_dim=128, rope_theta=1000000.0, max_position=131072,
    attention_bias=True))

# Falcon family — multi-query attention + parallel attn/MLP block → fallback
register(ModelConfig(
    name="falcon-7b", hidden_size=4544, num_layers=32, num_heads=71,
    num_kv_heads=1, intermediate_size=18176, vocab_size=65024, head_dim=64,
    rope_theta=10000.0, max_position=2048, tie_word_embeddings=True,
    runtime="transformers"))
_alias("falcon-7b-instruct", "falcon-7b")
register(ModelConfig(
    name="falcon-40b", hidden_size=8192, num_layers=60, num_heads=128,
    num_kv_heads=8, intermediate_size=32768, vocab_size=65024, head_dim=64,
    rope_theta=10000.0, max_position=2048, tie_word_embeddings=True,
    runtime="transformers"))
_alias("falcon-40b-instruct", "falcon-40b")

# Gemma-3 — native: sandwich norms, qk-norm, GeGLU, RMSNorm(1+w) folded
# at load, sqrt(H) embedding scale, 5-local+1-global sliding-window
# pattern with local-layer rope theta 10k.
register(ModelConfig(
    name="gemma-3-4b-instruct", hidden_size=2560, num_layers=34, num_heads=8,
    num_kv_heads=4, intermediate_size=10240, vocab_size=262208, head_dim=256,
    rope_theta=1000000.0, max_position=131072, tie_word_embeddings=True,
    sandwich_norms=True, qk_norm=True, rms_norm_offset=True,
    hidden_act="gelu_tanh", embed_scale=2560 ** 0.5, sliding_window=1024,
    sliding_window_pattern="interleaved:6", rope_theta_local=10000.0,
    rms_eps=1e-6))
register(ModelConfig(
    name="gemma-3-27b-instruct", hidden_size=5376, num_layers=62,
    num_heads=32, num_kv_heads=16, intermediate_size=21504,
    vocab_size=262208, head_dim=128, rope_theta=1000000.0,
    max_position=131072, tie_word_embeddings=True,
    sandwich_norms=True, qk_norm=True, rms_norm_offset=True,
    hidden_act="gelu_tanh", embed_scale=5376 ** 0.5, sliding_window=1024,
    sliding_window_pattern="interleaved:6", rope_theta_local=10000.0,
    rms_eps=1e-6, attn_scale=(5376 / 32) ** -0.5))

# gpt-oss MoE — native: learned attention sinks, alternating
# sliding-window layers, clamped-swiglu experts with biases,
# topk-then-softmax routing.
register(ModelConfig(
    name="gpt-oss-20b", hidden_size=2880, num_layers=24, num_heads=64,
    num_kv_heads=8, intermediate_size=2880, vocab_size=201088, head_dim=64,
    rope_theta=150000.0, max_position=131072, num_experts=32,
    num_experts_per_tok=4, moe_intermediate_size=2880,
    attn_sinks=True, sliding_window=128,
    sliding_window_pattern="interleaved:2", moe_act="swiglu_oai",
    moe_bias=True, moe_routing="topk_softmax", attention_bias=True))
register(ModelConfig(
    name="gpt-oss-120b", hidden_size=2880, num_layers=36, num_heads=64,
    num_kv_heads=8, intermediate_size=2880, vocab_size=201088, head_dim=64,
    rope_theta=150000.0, max_position=131072, num_experts=128,
    num_experts_per_tok=4, moe_intermediate_size=2880,
    attn_sinks=True, sliding_window=128,
    sliding_window_pattern="interleaved:2", moe_act="swiglu_oai",
    moe_bias=True, moe_routing="topk_softmax", attention_bias=True))

# DeepSeek V3/R1 — native MLA (models/mla.py + ops/csrc/
# mla_attention.hip): compressed 576-dim latent KV cache, absorbed
# decode, noaux_tc sigmoid routing with group-limited top-k, 1 shared
# expert, first 3 layers dense, yarn rope. 671B total params → TP=8
# (16 heads/rank == one decode-kernel head tile).
register(ModelConfig(
    name="deepseek-v3-0324", hidden_size=7168, num_layers=61, num_heads=128,
    num_kv_heads=1, intermediate_size=18432, vocab_size=129280,
    head_dim=192, rope_theta=10000.0, max_position=131072, num_experts=256,
    num_experts_per_tok=8, moe_intermediate_size=2048,
    kv_lora_rank=512, q_lora_rank=1536, qk_nope_head_dim=128,
    qk_rope_head_dim=64, v_head_dim=128,
    moe_routing="noaux_tc", moe_norm_topk=True, routed_scaling_factor=2.5,
    n_group=8, topk_group=4, n_shared_experts=1, first_k_dense=3,
    rope_scaling_type="yarn", rope_factor=40.0, rope_orig_max_position=4096,
    rope_mscale=1.0, rope_mscale_all_dim=1.0))
_alias("deepseek-r1-0528", "deepseek-v3-0324")

# DeepSeek V2-Lite — 15.7B MLA MoE that FITS one MI355X; direct q
# projection (no q-lora), 64 routed + 2 shared experts, softmax greedy
# routing without top-k renorm
register(ModelConfig(
    name="deepseek-v2-lite", hidden_size=2048, num_layers=27, num_heads=16,
    num_kv_heads=1, intermediate_size=10944, vocab_size=102400,
    head_dim=192, rope_theta=10000.0, max_position=163840, num_experts=64,
    num_experts_per_tok=6, moe_intermediate_size=1408,
    kv_lora_rank=512, q_lora_rank=0, qk_nope_head_dim=128,
    qk_rope_head_dim=64, v_head_dim=128,
    moe_routing="softmax_topk", moe_norm_topk=False,
    routed_scaling_factor=1.0, n_shared_experts=2, first_k_dense=1,
    rope_scaling_type="yarn", rope_factor=40.0, rope_orig_max_position=4096,
    rope_mscale=0.707, rope_mscale_all_dim=0.707, rms_eps=1e-6))
_alias("deepseek-v2-lite-chat", "deepseek-v2-lite")

# ---- tiny configs for tests ------------------------------------------------
register(ModelConfig(
    name="tiny-llama-test", hidden_size=256, num_layers=2, num_heads=4,
    num_kv_heads=2, intermediate_size=512, vocab_size=512, head_dim=64,
    rope_theta=10000.0, max_position=512))
register(ModelConfig(
    name="tiny-moe-test", hidden_size=256, num_layers=2, num_heads=4,
    num_kv_heads=2, intermediate_size=512, vocab_size=512, head_dim=64,
    rope_theta=10000.0, max_position=512, num_experts=4,
    num_experts_per_tok=2, moe_intermediate_size=256))
register(ModelConfig(
    name="tiny-phi2-test", hidden_size=256, num_layers=2, num_heads=4,
    num_kv_heads=4, intermediate_size=512, vocab_size=512, head_dim=64,
    rope_theta=10000.0, max_position=512, partial_rotary_factor=0.5,
    parallel_block=True, norm_type="layernorm", gated_mlp=False,
    hidden_act="gelu", attention_bias=True))
register(ModelConfig(
    name="tiny-gemma3-test", hidden_size=256, num_layers=4, num_heads=4,
    num_kv_heads=2, intermediate_size=512, vocab_size=512, head_dim=64,
    rope_theta=1000000.0, max_position=512, sandwich_norms=True,
    qk_norm=True, rms_norm_offset=True, hidden_act="gelu_tanh",
    embed_scale=16.0, sliding_window=32,
    sliding_window_pattern="interleaved:2", rope_theta_local=10000.0,
    tie_word_embeddings=True))
register(ModelConfig(
    # full deepseek feature set at kernel-real latent dims (r=512/rope=64
    # so the GPU decode kernel path is exercised): MLA with q-lora,
    # noaux_tc routing, shared expert, first layer dense, yarn rope
    name="tiny-deepseek-test", hidden_size=256, num_layers=2, num_heads=4,
    num_kv_heads=1, intermediate_size=512, vocab_size=512, head_dim=192,
    rope_theta=10000.0, max_position=512, num_experts=4,
    num_experts_per_tok=2, moe_intermediate_size=128,
    kv_lora_rank=512, q_lora_rank=64, qk_nope_head_dim=128,
    qk_rope_head_dim=64, v_head_dim=128,
    moe_routing="noaux_tc", moe_norm_topk=True, routed_scaling_factor=2.5,
    n_group=2, topk_group=1, n_shared_experts=1, first_k_dense=1,
    rope_scaling_type="yarn", rope_factor=8.0, rope_orig_max_position=64,
    rope_mscale=1.0, rope_mscale_all_dim=1.0))
register(ModelConfig(
    name="tiny-gptoss-test", hidden_size=256, num_layers=2, num_heads=4,
    num_kv_heads=2, intermediate_size=256, vocab_size=512, head_dim=64,
    rope_theta=150000.0, max_position=512, num_experts=4,
    num_experts_per_tok=2, moe_intermediate_size=256, attn_sinks=True,
    sliding_window=32, sliding_window_pattern="interleaved:2",
    moe_act="swiglu_oai", moe_bias=True, moe_routing="topk_softmax",
    attention_bias=True))
