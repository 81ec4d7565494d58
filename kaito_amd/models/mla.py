"""Multi-head Latent Attention (DeepSeek V2/V3/R1) — MI355X-native.

The reference serves the DeepSeek presets through vLLM's MLA path
(presets/workspace/models/supported_models.yaml: deepseek-v3-0324,
deepseek-r1-0528). This is the kaito_amd engine's equivalent, designed
for the CDNA4 memory system rather than translated:

  * The paged KV cache stores ONE compressed latent row per token —
    c_kv[kv_lora_rank] ‖ k_rope[qk_rope_head_dim] (576 bf16 for the
    DeepSeek family) — shared by EVERY attention head. At 128 heads
    that is a 64x cache-byte reduction vs MHA, which on a 288 GB
    MI355X means the whole 128k-token context window of many sequences
    stays resident.
  * Decode runs ABSORBED: q_nope is folded through W_uk into the
    latent space once per step (a small hipBLASLt bmm), and the decode
    kernel (ops/csrc/mla_attention.hip) streams the latent cache once
    per (seq, 16-head tile), staging each 16-token block through LDS so
    all 16 heads reuse every byte read from HBM. Scores ARE the
    absorbed dot (q_c·c_kv + q_pe·k_pe == q_nope·k_nope + rope term,
    exact linear algebra), and the V-accumulate output lives in the
    latent space too; W_uv is applied afterwards as a second bmm.
  * TP splits HEADS; the latent cache is replicated per rank (it is
    MQA-like — replicating 576 B/token beats sharding per-head K/V and
    re-gathering over xGMI). At deepseek-v3 scale (128 heads, TP=8)
    each rank holds 16 heads == exactly one kernel head-tile.
  * Prefill runs NON-absorbed: k_nope/v are decompressed from the
    fresh c_kv (bmm), q/k padded 192->256 and v 128->256 with zeros
    (exact — zero dims contribute nothing) through the existing
    D=256 prefill_attention MFMA kernel. MLA prompts are scheduled
    whole (scheduler.whole_prompt_only), so no context-attention
    (chunked suffix) variant is needed.

HF config fields covered: q_lora_rank (0 = direct q projection,
V2-Lite), kv_lora_rank, qk_nope_head_dim, qk_rope_head_dim, v_head_dim,
yarn rope scaling (see build_cos_sin_cache in llama.py).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from ..engine.config import ModelConfig
from ..parallel.state import get_state
from ..parallel.layers import ColumnParallelLinear, RowParallelLinear


def yarn_mscale(factor: float, mscale: float) -> float:
    if factor <= 1.0 or mscale <= 0.0:
        return 1.0
    return 0.1 * mscale * math.log(factor) + 1.0


def mla_softmax_scale(cfg: ModelConfig) -> float:
    """(nope+rope)^-0.5, yarn-adjusted (the HF deepseek convention:
    softmax_scale *= yarn_mscale(factor, mscale_all_dim)^2)."""
    scale = (cfg.qk_nope_head_dim + cfg.qk_rope_head_dim) ** -0.5
    if cfg.rope_scaling_type == "yarn" and cfg.rope_mscale_all_dim > 0:
        m = yarn_mscale(cfg.rope_factor, cfg.rope_mscale_all_dim)
        scale = scale * m * m
    return scale


class MLAAttention(nn.Module):
    """Drop-in for LlamaAttention when cfg.kv_lora_rank > 0. Same forward
    signature; kv_cache is the aliased (latent, latent) pair allocated by
    model_runner.profile_and_allocate_kv."""

    def __init__(self, cfg: ModelConfig, layer_idx: int = 0):
        super().__init__()
        tp = get_state().tp_size
        self.cfg = cfg
        h = cfg.hidden_size
        self.num_heads = cfg.num_heads // tp
        self.nope = cfg.qk_nope_head_dim
        self.rope = cfg.qk_rope_head_dim
        self.v_dim = cfg.v_head_dim
        self.r = cfg.kv_lora_rank
        self.qk_dim = self.nope + self.rope
        self.scale = cfg.attn_scale or mla_softmax_scale(cfg)
        self.window = 0

        if cfg.q_lora_rank > 0:
            # V3: x -> q_a [q_lora] -> rmsnorm -> q_b [H*(nope+rope)]
            self.q_a_proj = nn.Parameter(
                torch.empty(cfg.q_lora_rank, h, dtype=cfg.dtype),
                requires_grad=False)
            self.q_a_layernorm = nn.Parameter(
                torch.empty(cfg.q_lora_rank, dtype=cfg.dtype),
                requires_grad=False)
            self.q_b_proj = ColumnParallelLinear(
                cfg.q_lora_rank, cfg.num_heads * self.qk_dim,
                dtype=cfg.dtype)
        else:
            # V2-Lite: direct projection
            self.q_proj = ColumnParallelLinear(
                h, cfg.num_heads * self.qk_dim, dtype=cfg.dtype)

        # replicated: every rank computes the same latent row (and writes
        # the same cache), so decode needs NO collective before attention
        self.kv_a_proj_with_mqa = nn.Parameter(
            torch.empty(self.r + self.rope, h, dtype=cfg.dtype),
            requires_grad=False)
        self.kv_a_layernorm = nn.Parameter(
            torch.empty(self.r, dtype=cfg.dtype), requires_grad=False)

        # kv_b_proj [H*(nope+v), r] split into the absorbed operands:
        # w_kc [H_l, nope, r] (q absorption / k decompression) and
        # w_vc [H_l, r, v] (latent-space output -> per-head v)
        self.w_kc = nn.Parameter(
            torch.empty(self.num_heads, self.nope, self.r, dtype=cfg.dtype),
            requires_grad=False)
        self.w_vc = nn.Parameter(
            torch.empty(self.num_heads, self.r, self.v_dim, dtype=cfg.dtype),
            requires_grad=False)

        self.o_proj = RowParallelLinear(
            cfg.num_heads * self.v_dim, h, bias=False, dtype=cfg.dtype)
        self.o_proj.fuse_norm = not cfg.parallel_block \
            and not cfg.sandwich_norms and cfg.norm_type == "rmsnorm"

    def _project_q(self, x: torch.Tensor) -> torch.Tensor:
        if self.cfg.q_lora_rank > 0:
            qa = torch.nn.functional.linear(x, self.q_a_proj)
            qa = ops.rms_norm(qa, self.q_a_layernorm, self.cfg.rms_eps)
            q = self.q_b_proj(qa)
        else:
            q = self.q_proj(x)
        return q.view(-1, self.num_heads, self.qk_dim)

    def forward(self, x: torch.Tensor, positions: torch.Tensor,
                kv_cache: Optional[Tuple[torch.Tensor, torch.Tensor]],
                meta, cos_sin: torch.Tensor,
                cos_sin_local: Optional[torch.Tensor] = None) -> torch.Tensor:
        T = x.size(0)
        q = self._project_q(x)                          # [T, Hl, nope+rope]
        q_nope = q[..., :self.nope]
        q_pe = q[..., self.nope:].contiguous()          # [T, Hl, rope]

        kv_a = torch.nn.functional.linear(x, self.kv_a_proj_with_mqa)
        c_kv = ops.rms_norm(kv_a[:, :self.r].contiguous(),
                            self.kv_a_layernorm, self.cfg.rms_eps)
        k_pe = kv_a[:, self.r:].contiguous()            # [T, rope]

        q_pe, k_pe = ops.rotary_embedding(
            positions, q_pe.view(T, -1), k_pe, self.rope, cos_sin)
        q_pe = q_pe.view(T, self.num_heads, self.rope)

        if kv_cache is not None:
            cache = kv_cache[0]                    # [NB+1, BS, r+rope]
            if cache.is_cuda:
                # one fused scatter (skips -1 padding slots in-kernel);
                # replaces a where+cat+index_copy_ chain that cost ~7%
                # of a serving run (profiles/r02_mla_kernels.md)
                ops.mla_cache_write(cache, c_kv, k_pe, meta.slot_mapping)
            else:
                rows = cache.view(-1, cache.size(-1))
                slots = meta.slot_mapping
                slots = torch.where(
                    slots < 0, torch.full_like(slots, rows.size(0) - 1),
                    slots)
                rows.index_copy_(
                    0, slots,
                    torch.cat([c_kv, k_pe], dim=-1).to(cache.dtype))

        if meta.is_prefill:
            if meta.kv_lens is not None:
                # chunked-prefill continuation / prefix-cache restore:
                # suffix queries attend to the paged latent cache
                out = self._context(q_nope, q_pe, kv_cache[0], meta)
            else:
                out = self._prefill(q_nope, q_pe, c_kv, k_pe, meta)
        else:
            out = self._decode(q_nope, q_pe, kv_cache[0], meta)
        return self.o_proj(out.reshape(T, -1))

    # ---- absorbed decode over the latent cache --------------------------
    def _decode(self, q_nope, q_pe, cache, meta) -> torch.Tensor:
        T = q_nope.size(0)
        # q_c[h] = q_nope[h] @ w_kc[h]: [Hl, T, nope] x [Hl, nope, r].
        # strided views feed hipBLASLt batched GEMM directly (no
        # .contiguous() copies — profiles/r02_mla_kernels.md overhead)
        q_c = torch.bmm(q_nope.transpose(0, 1), self.w_kc)
        q_full = torch.cat(
            [q_c.transpose(0, 1), q_pe], dim=-1).contiguous()  # [T,Hl,r+rope]
        out_c = ops.mla_decode(q_full, cache, meta.block_tables,
                               meta.seq_lens, self.scale, self.r)
        # latent -> per-head v: [Hl, T, r] x [Hl, r, v]
        o = torch.bmm(out_c.transpose(0, 1), self.w_vc)
        return o.transpose(0, 1).reshape(T, -1)         # one copy at most

    # ---- absorbed suffix attention over the paged latent cache ----------
    def _context(self, q_nope, q_pe, cache, meta) -> torch.Tensor:
        """Chunked-prefill continuation (the GQA engine's
        context_attention analog). Runs in the ABSORBED space — no K/V
        decompression: gather each sequence's latent rows once, causal-
        mask the suffix, accumulate in latent space. Eager torch ops
        (prefill steps are eager; these chunks are the budget-split
        tail, not the hot path)."""
        BS = cache.size(1)
        DT = cache.size(2)
        q_c = torch.bmm(q_nope.transpose(0, 1), self.w_kc)  # [Hl, Tq, r]
        qf = torch.cat([q_c, q_pe.transpose(0, 1)], -1).float()
        outs = []
        cs = meta.cu_seqlens.tolist()
        for b in range(len(cs) - 1):
            s0, s1 = cs[b], cs[b + 1]
            n = s1 - s0
            L = int(meta.kv_lens[b])
            nb = (L + BS - 1) // BS
            rows = cache[meta.block_tables[b, :nb].long()]
            rows = rows.reshape(-1, DT)[:L].float()         # [L, DT]
            att = torch.einsum("hqd,ld->hql", qf[:, s0:s1], rows)
            att = att * self.scale
            qpos = torch.arange(L - n, L, device=att.device)
            mask = torch.arange(L, device=att.device).unsqueeze(0) \
                > qpos.unsqueeze(1)                         # [n, L]
            att = att.masked_fill(mask.unsqueeze(0), float("-inf"))
            p = torch.softmax(att, dim=-1)
            outs.append(torch.einsum("hql,lr->hqr", p, rows[:, :self.r]))
        out_c = torch.cat(outs, 1).to(self.w_vc.dtype)      # [Hl, Tq, r]
        o = torch.bmm(out_c, self.w_vc)                     # [Hl, Tq, v]
        return o.transpose(0, 1)

    # ---- non-absorbed whole-prompt prefill ------------------------------
    def _prefill(self, q_nope, q_pe, c_kv, k_pe, meta) -> torch.Tensor:
        T = q_nope.size(0)
        Hl = self.num_heads
        # decompress: k_nope [Hl, nope, T] = w_kc [Hl, nope, r] @ c_kv^T
        k_nope = torch.matmul(self.w_kc, c_kv.t().to(self.w_kc.dtype))
        k_nope = k_nope.permute(2, 0, 1)                # [T, Hl, nope]
        v = torch.matmul(c_kv.to(self.w_vc.dtype).unsqueeze(0), self.w_vc)
        v = v.transpose(0, 1)                           # [T, Hl, v]

        pad = 256                                       # D=256 MFMA kernel
        qf = torch.zeros(T, Hl, pad, dtype=q_nope.dtype,
                         device=q_nope.device)
        kf = torch.zeros_like(qf)
        vf = torch.zeros_like(qf)
        qf[..., :self.nope] = q_nope
        qf[..., self.nope:self.qk_dim] = q_pe
        kf[..., :self.nope] = k_nope
        kf[..., self.nope:self.qk_dim] = k_pe.unsqueeze(1)  # broadcast heads
        vf[..., :self.v_dim] = v
        out = ops.prefill_attention(qf, kf, vf, meta.cu_seqlens, self.scale,
                                    meta.max_seqlen)
        return out[..., :self.v_dim].contiguous()       # [T, Hl, v]
