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


def get_model_config(name: str) -> ModelConfig:
    key = name.lower()
    if key not in _REGISTRY:
        raise KeyError(f"unknown model preset: {name}; known: {sorted(_REGISTRY)}")
    return _REGISTRY[key]


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
