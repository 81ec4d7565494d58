"""Llama-architecture causal LM built on kaito_amd's gfx950 HIP ops.

Covers Llama-3 8B/70B (BASELINE configs #2/#3), Mistral, Qwen-dense and the
Phi-4-mini class (partial rotary). GEMMs go through hipBLASLt
(torch.nn.functional.linear); everything between GEMMs is our fused HIP
kernels (rmsnorm, rope, silu_mul, paged/prefill attention).

Reference parity: this is the engine-side replacement for the vLLM model
executor that KAITO launches via presets/workspace/inference/vllm/
inference_api.py.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from ..engine.config import ModelConfig
from ..parallel.state import get_state
from ..parallel.layers import (ColumnParallelLinear, RowParallelLinear,
                               VocabParallelEmbedding)


@dataclass
class AttnMetadata:
    """Per-batch attention metadata. Exactly one of (prefill, decode)."""
    is_prefill: bool
    slot_mapping: torch.Tensor                  # [T] int64
    # prefill:
    cu_seqlens: Optional[torch.Tensor] = None   # [B+1] int32 (q suffix lens)
    max_seqlen: int = 0
    # decode:
    block_tables: Optional[torch.Tensor] = None  # [T, max_blocks] int32
    seq_lens: Optional[torch.Tensor] = None      # [T] int32
    # context-prefill (chunked prefill / prefix-cache continuation):
    # q covers only suffix tokens; K/V read from the paged cache.
    kv_lens: Optional[torch.Tensor] = None       # [B] int32 total ctx len


def build_cos_sin_cache(cfg: ModelConfig, device, max_pos: Optional[int] = None,
                        theta: Optional[float] = None) -> torch.Tensor:
    """[max_pos, rot_dim] f32 = [cos | sin] table. Supports YaRN scaling
    (DeepSeek long-context: NTK-by-parts frequency interpolation with the
    beta_fast/beta_slow ramp + mscale on cos/sin)."""
    rot = cfg.rotary_dim
    max_pos = max_pos or cfg.max_position
    theta = theta or cfg.rope_theta
    idx = torch.arange(0, rot, 2, dtype=torch.float64, device=device)
    inv = 1.0 / (theta ** (idx / rot))
    mscale = 1.0
    if cfg.rope_scaling_type == "yarn" and cfg.rope_factor > 1.0:
        from .mla import yarn_mscale
        orig = cfg.rope_orig_max_position or cfg.max_position
        inv_interp = inv / cfg.rope_factor

        def corr_dim(n_rot: float) -> float:
            return (rot * math.log(orig / (n_rot * 2 * math.pi))
                    / (2 * math.log(theta)))

        low = max(math.floor(corr_dim(cfg.rope_beta_fast)), 0)
        high = min(math.ceil(corr_dim(cfg.rope_beta_slow)), rot // 2 - 1)
        ramp = ((torch.arange(rot // 2, dtype=torch.float64, device=device)
                 - low) / max(high - low, 1e-3)).clamp(0.0, 1.0)
        extrap_mask = 1.0 - ramp        # low dims extrapolate, high interp
        inv = inv * extrap_mask + inv_interp * (1.0 - extrap_mask)
        # HF deepseek: cos/sin scaled by mscale(factor, mscale) /
        # mscale(factor, mscale_all_dim) (net 1.0 when both equal, e.g.
        # V3); the all_dim part goes into the softmax scale (mla.py)
        mscale = yarn_mscale(cfg.rope_factor, cfg.rope_mscale)
        if cfg.rope_mscale_all_dim > 0:
            mscale /= yarn_mscale(cfg.rope_factor, cfg.rope_mscale_all_dim)
    t = torch.arange(max_pos, dtype=torch.float64, device=device)
    freqs = torch.outer(t, inv)
    return (torch.cat([freqs.cos(), freqs.sin()], dim=-1)
            * mscale).float().contiguous()


class LlamaAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int = 0):
        super().__init__()
        tp = get_state().tp_size
        self.cfg = cfg
        self.head_dim = cfg.head_dim
        self.num_heads = cfg.num_heads // tp
        self.num_kv_heads = max(cfg.num_kv_heads // tp, 1)
        self.scale = cfg.attn_scale or 1.0 / math.sqrt(self.head_dim)
        # sliding-window layers (gemma-3 / gpt-oss interleaved patterns)
        self.window = cfg.layer_sliding_window(layer_idx)
        # gemma-3: local (sliding) layers use a different rope theta
        self.local_rope = cfg.rope_theta_local > 0 and self.window > 0
        h = cfg.hidden_size
        self.qkv_proj = ColumnParallelLinear(
            h, (cfg.num_heads + 2 * cfg.num_kv_heads) * cfg.head_dim,
            bias=cfg.attention_bias, dtype=cfg.dtype)
        self.o_proj = RowParallelLinear(
            cfg.num_heads * cfg.head_dim, h, bias=False, dtype=cfg.dtype)
        # o_proj feeds post_attention_layernorm: with an active one-shot
        # group its ring all-reduce is deferred into the fused
        # allreduce+add+RMSNorm kernel (parallel/one_shot.py); only the
        # standard pre-norm block structure has that call site
        self.o_proj.fuse_norm = (not cfg.parallel_block
                                 and not cfg.sandwich_norms
                                 and cfg.norm_type == "rmsnorm")
        self.q_size = self.num_heads * self.head_dim
        self.kv_size = self.num_kv_heads * self.head_dim
        if cfg.qk_norm:   # per-head RMSNorm before RoPE (gemma-3, qwen3)
            self.q_norm = nn.Parameter(
                torch.empty(cfg.head_dim, dtype=cfg.dtype),
                requires_grad=False)
            self.k_norm = nn.Parameter(
                torch.empty(cfg.head_dim, dtype=cfg.dtype),
                requires_grad=False)
        if cfg.attn_sinks:  # learned softmax sinks (gpt-oss)
            self.sinks = nn.Parameter(
                torch.empty(self.num_heads, dtype=torch.float32),
                requires_grad=False)

    def forward(self, x: torch.Tensor, positions: torch.Tensor,
                kv_cache: Optional[Tuple[torch.Tensor, torch.Tensor]],
                meta: AttnMetadata, cos_sin: torch.Tensor,
                cos_sin_local: Optional[torch.Tensor] = None) -> torch.Tensor:
        qkv = self.qkv_proj(x)
        # strided views into the fused qkv buffer — the HIP kernels take row
        # strides, so no .contiguous() copies on the hot path.
        q, k, v = qkv.split([self.q_size, self.kv_size, self.kv_size], dim=-1)
        if self.cfg.qk_norm:
            q = ops.rms_norm(
                q.reshape(-1, self.head_dim).contiguous(), self.q_norm,
                self.cfg.rms_eps).reshape(q.shape[0], -1)
            k = ops.rms_norm(
                k.reshape(-1, self.head_dim).contiguous(), self.k_norm,
                self.cfg.rms_eps).reshape(k.shape[0], -1)
            # k is now a fresh contiguous tensor; reshape_and_cache needs
            # k/v with EQUAL row strides, so v must match
            v = v.contiguous()
        cs = cos_sin_local if self.local_rope else cos_sin
        q, k = ops.rotary_embedding(positions, q, k, self.head_dim, cs)
        T = x.size(0)
        qh = q.unflatten(-1, (self.num_heads, self.head_dim))
        kh = k.unflatten(-1, (self.num_kv_heads, self.head_dim))
        vh = v.unflatten(-1, (self.num_kv_heads, self.head_dim))
        if kv_cache is not None:
            ops.reshape_and_cache(kh, vh, kv_cache[0], kv_cache[1],
                                  meta.slot_mapping)
        sinks = getattr(self, "sinks", None)
        if meta.is_prefill:
            if meta.kv_lens is not None:
                # suffix-query attention over the paged cache (the suffix
                # K/V was just written by reshape_and_cache above)
                out = ops.context_attention(qh, kv_cache[0], kv_cache[1],
                                            meta.cu_seqlens, meta.kv_lens,
                                            meta.block_tables, self.scale,
                                            self.window, sinks)
            else:
                out = ops.prefill_attention(qh, kh, vh, meta.cu_seqlens,
                                            self.scale, meta.max_seqlen,
                                            self.window, sinks)
        else:
            out = ops.paged_attention(qh, kv_cache[0], kv_cache[1],
                                      meta.block_tables, meta.seq_lens,
                                      self.scale, self.window, sinks)
        return self.o_proj(out.reshape(T, -1))


class LlamaMLP(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.cfg = cfg
        # num_experts>0 ⇒ the model mixes dense + MoE layers (deepseek
        # first_k_dense); the next layer's norm assumes a NON-deferred
        # producer there (LlamaDecoderLayer deferred_producer flag), so
        # dense layers in MoE models must not defer either
        fuse = (not cfg.parallel_block and not cfg.sandwich_norms
                and cfg.norm_type == "rmsnorm" and cfg.num_experts == 0)
        if cfg.gated_mlp:
            self.gate_up_proj = ColumnParallelLinear(
                cfg.hidden_size, 2 * cfg.intermediate_size, dtype=cfg.dtype)
            self.down_proj = RowParallelLinear(
                cfg.intermediate_size, cfg.hidden_size, dtype=cfg.dtype)
        else:
            # ungated (phi-2/falcon): fc1 → act → fc2, bias follows the
            # model's attention-bias convention
            self.up_proj = ColumnParallelLinear(
                cfg.hidden_size, cfg.intermediate_size,
                bias=cfg.attention_bias, dtype=cfg.dtype)
            self.down_proj = RowParallelLinear(
                cfg.intermediate_size, cfg.hidden_size,
                bias=cfg.attention_bias, dtype=cfg.dtype)
        # down_proj feeds the next input_layernorm / final norm
        self.down_proj.fuse_norm = fuse
        tp = get_state().tp_size
        self.inter_per_rank = cfg.intermediate_size // tp

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not self.cfg.gated_mlp:
            return self.down_proj(ops.gelu(self.up_proj(x)))
        gu = self.gate_up_proj(x)
        act = ops.gelu_and_mul(gu) if self.cfg.hidden_act == "gelu_tanh" \
            else ops.silu_and_mul(gu)
        return self.down_proj(act)


def _tp_fused_add_rms_norm(hidden, residual, weight, eps: float,
                           deferred_producer: bool):
    """fused_add_rms_norm that ALSO performs the TP reduction when the
    producing RowParallelLinear deferred it to the one-shot group
    (allreduce + residual-add + RMSNorm in ONE kernel over xGMI peer
    staging — the fusion the reference disables on NVIDIA,
    interface.go:439-446). Falls back to the already-reduced plain fused
    norm when no group is active or the batch exceeds the staging
    window (the RowParallel layer makes the same `defer` decision, so
    the paths always agree)."""
    if deferred_producer and get_state().tp_size > 1:
        from ..parallel import one_shot
        grp = one_shot.defer(hidden.reshape(-1, hidden.shape[-1]).shape[0])
        if grp is not None:
            return grp.allreduce_add_rmsnorm(hidden, residual, weight, eps)
    return ops.fused_add_rms_norm(hidden, residual, weight, eps)


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int = 0):
        super().__init__()
        self.cfg = cfg
        if cfg.is_mla:
            from .mla import MLAAttention
            self.self_attn = MLAAttention(cfg, layer_idx)
        else:
            self.self_attn = LlamaAttention(cfg, layer_idx)
        # deepseek: the first k layers are dense even in MoE models
        if cfg.num_experts > 0 and layer_idx >= cfg.first_k_dense:
            from .moe import MoEMLP
            self.mlp = MoEMLP(cfg)
        else:
            self.mlp = LlamaMLP(cfg)

        def norm_w():
            return nn.Parameter(torch.empty(cfg.hidden_size, dtype=cfg.dtype),
                                requires_grad=False)

        self.input_layernorm = norm_w()
        if not cfg.parallel_block:   # phi-2 shares one norm per layer
            self.post_attention_layernorm = norm_w()
        if cfg.sandwich_norms:       # gemma-2/3 extra output norms
            self.pre_feedforward_layernorm = norm_w()
            self.post_feedforward_layernorm = norm_w()
        if cfg.norm_type == "layernorm":
            self.input_layernorm_bias = norm_w()
            if not cfg.parallel_block:
                self.post_attention_layernorm_bias = norm_w()

    def _norm(self, x, which: str):
        w = getattr(self, which)
        if self.cfg.norm_type == "layernorm":
            b = getattr(self, which + "_bias", None)
            return ops.layer_norm(x, w, b, self.cfg.rms_eps)
        return ops.rms_norm(x, w, self.cfg.rms_eps)

    def forward(self, hidden, residual, positions, kv_cache, meta, cos_sin,
                cos_sin_local=None):
        cfg = self.cfg
        if cfg.parallel_block:
            # phi-2/falcon: one shared norm; attn and mlp read the same
            # normed input; stream += attn_out + mlp_out
            stream = hidden if residual is None else \
                (hidden.float() + residual.float()).to(hidden.dtype)
            x = self._norm(stream, "input_layernorm")
            a = self.self_attn(x, positions, kv_cache, meta, cos_sin,
                               cos_sin_local)
            m = self.mlp(x)
            return (a.float() + m.float()).to(hidden.dtype), stream
        if cfg.sandwich_norms:
            # gemma-2/3: pre+post norms around BOTH sub-blocks (norms on
            # the sub-block OUTPUT before the residual add)
            stream = hidden if residual is None else \
                (hidden.float() + residual.float()).to(hidden.dtype)
            x = self._norm(stream, "input_layernorm")
            a = self.self_attn(x, positions, kv_cache, meta, cos_sin,
                               cos_sin_local)
            a = self._norm(a, "post_attention_layernorm")
            stream = (stream.float() + a.float()).to(hidden.dtype)
            y = self._norm(stream, "pre_feedforward_layernorm")
            m = self.mlp(y)
            m = self._norm(m, "post_feedforward_layernorm")
            return m, stream
        if residual is None:
            residual = hidden
            hidden = self._norm(hidden, "input_layernorm")
        elif cfg.norm_type == "layernorm":
            hidden, residual = ops.fused_add_layer_norm(
                hidden, residual, self.input_layernorm,
                self.input_layernorm_bias, cfg.rms_eps)
        else:
            # `hidden` here is the PREVIOUS layer's mlp output — a
            # deferred TP partial when a one-shot group is active and the
            # mlp is the dense LlamaMLP (MoE's reduce is not deferred)
            hidden, residual = _tp_fused_add_rms_norm(
                hidden, residual, self.input_layernorm, cfg.rms_eps,
                deferred_producer=cfg.num_experts == 0)
        hidden = self.self_attn(hidden, positions, kv_cache, meta, cos_sin,
                                cos_sin_local)
        if cfg.norm_type == "layernorm":
            hidden, residual = ops.fused_add_layer_norm(
                hidden, residual, self.post_attention_layernorm,
                self.post_attention_layernorm_bias, cfg.rms_eps)
        else:
            hidden, residual = _tp_fused_add_rms_norm(
                hidden, residual, self.post_attention_layernorm, cfg.rms_eps,
                deferred_producer=True)   # o_proj always defer-capable
        hidden = self.mlp(hidden)
        return hidden, residual


class LlamaForCausalLM(nn.Module):
    """Pipeline-aware: with pp_size>1 each stage builds only its layer
    slice; stage 0 owns the embedding, the last stage owns norm + lm_head.
    forward() takes token ids on stage 0 and hidden states elsewhere."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.cfg = cfg
        st = get_state()
        self.pp_rank, self.pp_size = st.pp_rank, st.pp_size
        per = (cfg.num_layers + self.pp_size - 1) // self.pp_size
        self.layer_start = self.pp_rank * per
        self.layer_end = min(cfg.num_layers, self.layer_start + per)
        self.is_first = st.is_first_stage
        self.is_last = st.is_last_stage
        self.embed_tokens = VocabParallelEmbedding(
            cfg.vocab_size, cfg.hidden_size, dtype=cfg.dtype) \
            if (self.is_first or cfg.tie_word_embeddings) else None
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg, self.layer_start + i)
             for i in range(self.layer_end - self.layer_start)])
        self.norm = nn.Parameter(
            torch.empty(cfg.hidden_size, dtype=cfg.dtype),
            requires_grad=False) if self.is_last else None
        self.norm_bias = nn.Parameter(
            torch.empty(cfg.hidden_size, dtype=cfg.dtype),
            requires_grad=False) \
            if (self.is_last and cfg.norm_type == "layernorm") else None
        self.lm_head = None
        if self.is_last and not cfg.tie_word_embeddings:
            self.lm_head = ColumnParallelLinear(
                cfg.hidden_size, cfg.vocab_size, dtype=cfg.dtype)
        self.register_buffer("cos_sin_cache", torch.empty(0), persistent=False)
        self.cos_sin_cache_local = None

    @property
    def num_local_layers(self) -> int:
        return len(self.layers)

    def init_rope(self, device, max_pos: Optional[int] = None):
        self.cos_sin_cache = build_cos_sin_cache(self.cfg, device, max_pos)
        if self.cfg.rope_theta_local > 0:   # gemma-3 local-layer rope
            self.cos_sin_cache_local = build_cos_sin_cache(
                self.cfg, device, max_pos, theta=self.cfg.rope_theta_local)
        else:
            self.cos_sin_cache_local = None

    def forward(self, input_ids: torch.Tensor, positions: torch.Tensor,
                kv_caches: Optional[List[Tuple[torch.Tensor, torch.Tensor]]],
                meta: AttnMetadata,
                hidden_in: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Stage 0: embeds input_ids; later stages take hidden_in. Returns
        the stage output hidden (+ residual folded in): intermediate stages
        return the value to SEND; the last stage returns final normed
        hidden."""
        from ..parallel import one_shot
        grp = one_shot.active()
        if grp is not None:
            grp.begin_step()   # deterministic staging flip for hipGraphs
        if self.is_first:
            hidden = self.embed_tokens(input_ids)
            if self.cfg.embed_scale != 1.0:   # gemma: sqrt(hidden_size)
                hidden = (hidden.float() * self.cfg.embed_scale).to(
                    hidden.dtype)
            residual = None
        else:
            assert hidden_in is not None, "non-first PP stage needs hidden_in"
            hidden = hidden_in
            residual = None
        for i, layer in enumerate(self.layers):
            kv = kv_caches[i] if kv_caches is not None else None
            hidden, residual = layer(hidden, residual, positions, kv, meta,
                                     self.cos_sin_cache,
                                     self.cos_sin_cache_local)
        if self.is_last:
            if self.cfg.norm_type == "layernorm":
                hidden, _ = ops.fused_add_layer_norm(
                    hidden, residual, self.norm, self.norm_bias,
                    self.cfg.rms_eps)
            else:
                hidden, _ = _tp_fused_add_rms_norm(
                    hidden, residual, self.norm, self.cfg.rms_eps,
                    deferred_producer=(self.cfg.num_experts == 0
                                       and not self.cfg.parallel_block
                                       and not self.cfg.sandwich_norms))
            return hidden
        # fold the residual stream so one tensor crosses the stage boundary
        return (hidden.float() + residual.float()).to(hidden.dtype)

    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        from ..parallel.state import tp_all_gather
        if self.lm_head is not None:
            logits = self.lm_head(hidden)
        else:
            logits = torch.nn.functional.linear(hidden, self.embed_tokens.weight)
        logits = tp_all_gather(logits, dim=-1)
        cap = self.cfg.final_logit_softcap
        if cap > 0:   # gemma-2 logit soft-capping
            logits = (torch.tanh(logits.float() / cap) * cap).to(logits.dtype)
        return logits

    @torch.no_grad()
    def random_init(self, seed: int = 0):
        """Random-init weights of the right architecture (no network; bench
        contract: synthetic data / random weights). Generates on the params'
        device (device-side for 8B+ models; CPU init would take minutes)."""
        dev = next(self.parameters()).device
        gen = torch.Generator(device=dev).manual_seed(seed)
        for name, p in self.named_parameters():
            last = name.split(".")[-1]
            if last.endswith("_bias") or last == "norm_bias":
                p.zero_()
            elif last == "sinks":
                tmp = torch.empty(p.shape, dtype=torch.float32, device=dev)
                tmp.normal_(0, 0.5, generator=gen)
                p.copy_(tmp.to(p.dtype))
            elif ("layernorm" in name or name == "norm"
                    or last in ("q_norm", "k_norm")):
                p.fill_(1.0)
            else:
                std = 0.02 if "embed" in name or "lm_head" in name else \
                    0.02 / math.sqrt(2 * self.cfg.num_layers)
                tmp = torch.empty(p.shape, dtype=torch.float32, device=dev)
                tmp.normal_(0, std, generator=gen)
                p.copy_(tmp.to(p.dtype))
        return self
