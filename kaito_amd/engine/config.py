"""Engine + model configuration.

ModelConfig covers the llama-architecture family (Llama-3 8B/70B, Phi-4-mini
class, Mistral, Qwen dense) — the architectures behind the reference's
headline benchmarks (Llama-3-8B TP=1 / 70B TP=8,
Phi-4-mini serving CSVs).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

import torch


@dataclass
class ModelConfig:
    name: str = "llama-3-8b"
    hidden_size: int = 4096
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    intermediate_size: int = 14336
    vocab_size: int = 128256
    head_dim: int = 128
    rope_theta: float = 500000.0
    max_position: int = 8192
    rms_eps: float = 1e-5
    tie_word_embeddings: bool = False
    dtype: torch.dtype = torch.bfloat16
    # partial-rotary models (phi family): fraction of head_dim rotated
    partial_rotary_factor: float = 1.0
    # attention bias (qwen1/2 style)
    attention_bias: bool = False
    # ---- architecture variants (gemma / phi-2 / gpt-oss native support) ----
    # MLP activation: "silu" (llama SwiGLU) | "gelu_tanh" (gemma GeGLU) |
    # "gelu" (phi-2 plain-GELU, UNGATED fc1→act→fc2 MLP)
    hidden_act: str = "silu"
    # ungated MLP (phi-2/falcon): fc1 [I,H] → act → fc2 [H,I]
    gated_mlp: bool = True
    # norm layer: "rmsnorm" | "layernorm" (phi-2/falcon, with bias)
    norm_type: str = "rmsnorm"
    # gemma stores RMSNorm weights as w with y = x*(1+w); folded to (1+w)
    # at load/init time so the kernel is unchanged
    rms_norm_offset: bool = False
    # gemma scales embeddings by sqrt(hidden_size)
    embed_scale: float = 1.0
    # phi-2/falcon parallel block: h += attn(ln(x)) + mlp(ln(x))
    parallel_block: bool = False
    # gemma-2/3 sandwich norms: extra pre/post feedforward + post-attn norms
    sandwich_norms: bool = False
    # per-head RMSNorm on q/k before RoPE (gemma-3, qwen3)
    qk_norm: bool = False
    # sliding-window attention: window size (0 = full); layer pattern
    # "all" | "interleaved:N" (1 global every N layers — gemma-3 style,
    # global layers are (i+1) % N == 0)
    sliding_window: int = 0
    sliding_window_pattern: str = "all"
    # rope theta for LOCAL (sliding) layers when it differs (gemma-3:
    # local 10k, global rope_theta); 0 → use rope_theta everywhere
    rope_theta_local: float = 0.0
    # attention sinks (gpt-oss): learned per-head logit folded into the
    # softmax denominator
    attn_sinks: bool = False
    # final logit soft-capping (gemma-2): tanh cap value, 0 = off
    final_logit_softcap: float = 0.0
    # mixture-of-experts (mixtral / qwen-moe / gpt-oss class); 0 = dense
    num_experts: int = 0
    num_experts_per_tok: int = 2
    moe_intermediate_size: int = 0       # per-expert FFN width (0 → dense I)
    # expert activation: "silu" (SwiGLU) | "swiglu_oai" (gpt-oss clamped
    # swiglu: (up+1) * gate*sigmoid(1.702*gate), clamp ±7)
    moe_act: str = "silu"
    moe_bias: bool = False               # per-expert gate_up/down biases
    # routing: "softmax_topk" (mixtral: softmax→topk→renorm) |
    # "topk_softmax" (gpt-oss: topk logits→softmax over the k)
    moe_routing: str = "softmax_topk"
    # attention scale override (padded-head models: phi-2 pads 80→128 for
    # the D∈{64,128,256} kernels but keeps 80^-0.5 scaling; zero-padded
    # dims contribute nothing, so the math is exact)
    attn_scale: Optional[float] = None
    # ---- MLA (multi-head latent attention — DeepSeek V2/V3/R1) ----------
    # kv_lora_rank > 0 enables MLA: the paged cache stores ONE compressed
    # row per token (c_kv[kv_lora_rank] ‖ k_rope[qk_rope_head_dim]) shared
    # by every head; decode runs in the absorbed space (q_nope·W_uk
    # folded into q) over that latent cache (ops/csrc/mla_attention.hip)
    kv_lora_rank: int = 0
    q_lora_rank: int = 0                 # 0 = direct q projection (V2-Lite)
    qk_nope_head_dim: int = 128
    qk_rope_head_dim: int = 64
    v_head_dim: int = 128
    # ---- DeepSeek MoE extensions ----------------------------------------
    # routing adds "noaux_tc" (V3: sigmoid scores + e_score_correction_bias
    # for selection, group-limited top-k, ORIGINAL sigmoid scores as
    # weights) to softmax_topk/topk_softmax
    moe_norm_topk: bool = True           # renormalize selected weights
    routed_scaling_factor: float = 1.0   # V3: 2.5
    n_group: int = 1                     # expert groups (V3: 8)
    topk_group: int = 1                  # groups kept per token (V3: 4)
    n_shared_experts: int = 0            # always-on experts (width = n*moe_ie)
    first_k_dense: int = 0               # leading dense (non-MoE) layers
    # ---- YaRN rope scaling (DeepSeek long-context) ----------------------
    rope_scaling_type: str = ""          # "" | "yarn"
    rope_factor: float = 1.0
    rope_orig_max_position: int = 0      # original_max_position_embeddings
    rope_beta_fast: float = 32.0
    rope_beta_slow: float = 1.0
    rope_mscale: float = 1.0
    rope_mscale_all_dim: float = 0.0
    # weight quantization: "" (bf16) | "w4a16"/"awq" (4-bit group-quantized
    # linears via the HIP GEMV / dequant+MFMA kernels)
    quant_method: str = ""
    # serving runtime: "native" → HIP engine (llama-architecture family);
    # "transformers" → fallback runtime (reference: vLLM vs text-generation
    # runtime split, supported_models.yaml `runtime:` field)
    runtime: str = "native"

    @classmethod
    def from_hf_config(cls, cfg: dict, name: str = "") -> "ModelConfig":
        """Build from a HuggingFace config.json dict (the reference's preset
        generator derives BytesPerToken/size the same way,
        presets/workspace/generator/generator.go:389,660)."""
        h = cfg.get("hidden_size", 4096)
        heads = cfg.get("num_attention_heads", 32)
        arch = (cfg.get("architectures") or [""])[0]
        native = any(a in arch for a in (
            "Llama", "Mistral", "Qwen2", "Phi3", "Phi4", "Mixtral"))
        return cls(
            name=name or cfg.get("_name_or_path", "custom"),
            hidden_size=h,
            num_layers=cfg.get("num_hidden_layers", 32),
            num_heads=heads,
            num_kv_heads=cfg.get("num_key_value_heads", heads),
            intermediate_size=cfg.get("intermediate_size", 4 * h),
            vocab_size=cfg.get("vocab_size", 32000),
            head_dim=cfg.get("head_dim", h // heads),
            rope_theta=cfg.get("rope_theta", 10000.0),
            max_position=cfg.get("max_position_embeddings", 8192),
            rms_eps=cfg.get("rms_norm_eps", 1e-5),
            tie_word_embeddings=cfg.get("tie_word_embeddings", False),
            partial_rotary_factor=cfg.get("partial_rotary_factor", 1.0),
            attention_bias=cfg.get("attention_bias", False),
            num_experts=cfg.get("num_local_experts",
                                cfg.get("n_routed_experts", 0)) or 0,
            num_experts_per_tok=cfg.get("num_experts_per_tok", 2),
            moe_intermediate_size=cfg.get("moe_intermediate_size", 0),
            runtime="native" if native else "transformers",
        )

    @property
    def rotary_dim(self) -> int:
        if self.is_mla:          # rope applies to the decoupled rope dims
            return self.qk_rope_head_dim
        r = int(self.head_dim * self.partial_rotary_factor)
        return r - (r % 2)

    def layer_sliding_window(self, layer_idx: int) -> int:
        """Effective attention window for a layer (0 = full). Pattern
        "interleaved:N": one global layer every N (layers with
        (i+1) % N == 0 are global — gemma-3 is interleaved:6, gpt-oss
        alternates as interleaved:2)."""
        if self.sliding_window <= 0:
            return 0
        if self.sliding_window_pattern == "all":
            return self.sliding_window
        if self.sliding_window_pattern.startswith("interleaved:"):
            n = int(self.sliding_window_pattern.split(":", 1)[1])
            return 0 if (layer_idx + 1) % n == 0 else self.sliding_window
        raise ValueError(self.sliding_window_pattern)

    @property
    def is_mla(self) -> bool:
        return self.kv_lora_rank > 0

    @property
    def kv_cache_row(self) -> int:
        """Per-token latent cache width for MLA models."""
        return self.kv_lora_rank + self.qk_rope_head_dim

    def kv_bytes_per_token(self, tp_size: int = 1) -> int:
        """Per-token KV cache bytes across all layers (per TP rank)."""
        if self.is_mla:
            # one compressed latent row shared by all heads, REPLICATED
            # across TP ranks (MQA-like; the xGMI win is skipping the
            # per-head cache entirely)
            return self.num_layers * self.kv_cache_row * 2
        kvh = max(self.num_kv_heads // tp_size, 1)
        return 2 * self.num_layers * kvh * self.head_dim * 2  # k+v, bf16

    def param_bytes(self, tp_size: int = 1) -> int:
        h, i, v = self.hidden_size, self.intermediate_size, self.vocab_size
        qkv = h * (self.num_heads + 2 * self.num_kv_heads) * self.head_dim
        o = self.num_heads * self.head_dim * h
        if self.num_experts > 0:
            ie = self.moe_intermediate_size or i
            mlp = self.num_experts * 3 * h * ie + h * self.num_experts
        else:
            mlp = 3 * h * i
        per_layer = (qkv + o) // tp_size + mlp // tp_size + 2 * h
        embed = v * h * (1 if self.tie_word_embeddings else 2)
        return 2 * (self.num_layers * per_layer + embed + h)


@dataclass
class EngineConfig:
    model: ModelConfig = field(default_factory=ModelConfig)
    block_size: int = 16
    # KV cache dtype: "auto"/"bf16" | "fp8" (OCP e4m3, GPU only —
    # halves attention bytes and doubles KV capacity; vLLM's
    # --kv-cache-dtype fp8 analog, unscaled)
    kv_cache_dtype: str = "auto"
    max_num_seqs: int = 1024
    max_num_batched_tokens: int = 8192      # prefill token budget per step
    max_model_len: Optional[int] = None     # None → "auto": fit KV budget
    gpu_memory_utilization: float = 0.90
    num_gpu_blocks: Optional[int] = None    # None → probe free VRAM
    tensor_parallel_size: int = 1
    enforce_eager: bool = False             # disable hipGraph decode capture
    enable_lora: bool = False
    max_loras: int = 8
    max_lora_rank: int = 64
    enable_prefix_caching: bool = True      # block-hash APC (zero-copy)
    kv_offload: bool = False                # LMCache-style host KV cache
    kv_offload_bytes: Optional[int] = None  # None → 0.5 * available RAM
    # mixed steps: decode every step, with a bounded prefill chunk run
    # fire-and-forget on a side HIP stream. MEASURED (profiles/
    # r01_decode_profile.md): a throughput LOSS at the saturated bs=1024
    # headline point — decode already fills the device, so overlap creates
    # no capacity and the per-step eager prefill launches cost host time.
    # Off by default; the right operating point for it is latency-lean
    # small-batch serving (prefill bursts stall decode there).
    enable_mixed_batch: bool = False
    mixed_prefill_tokens: int = 2048        # per-step overlap prefill budget
    # fused one-shot allreduce+add+RMSNorm over xGMI for TP decode
    # (parallel/one_shot.py). None → auto: on when tp>1 and pp==1 (the
    # hipIpc peer mapping needs all TP ranks on one node). Batches larger
    # than max_num_seqs fall back to the RCCL ring.
    enable_one_shot_allreduce: Optional[bool] = None
    device: str = "cuda"
    seed: int = 0
    # decode graph buckets (batch sizes to capture)
    graph_batch_sizes: tuple = (1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128,
                                192, 256, 320, 384, 448, 512, 640, 768,
                                896, 1024, 1280, 1536, 2048)

    def max_blocks_per_seq(self, max_len: int) -> int:
        return (max_len + self.block_size - 1) // self.block_size
