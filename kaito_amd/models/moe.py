"""Mixture-of-experts FFN (Mixtral / Qwen-MoE / gpt-oss class).

Replaces vLLM's fused-MoE CUDA path for the MoE presets in the reference
catalog (gpt-oss-120b / Qwen MoE presets in
supported_models.yaml).

Round 2: the GPU path is FUSED and host-sync-free — tokens are sorted by
expert on the device and two HIP grouped-GEMM kernels (ops/csrc/moe.hip)
consume (sorted_ids, offsets) directly: gather→gate/up GEMM→silu·mul,
then down GEMM→weighted scatter-add. Worst-case tile grids make every
shape static, so MoE decode steps capture into hipGraphs (the round-1
per-expert torch loop read expert counts on the host each layer, forcing
eager decode — docs/ROADMAP.md #3).

Parallelism at tp>1:
  * EP (expert parallel) when num_experts % tp == 0: each rank owns
    E/tp experts; non-local tokens contribute nothing locally and the
    rank outputs are summed by the existing TP all-reduce.
  * TP fallback (IE sharding) otherwise: every rank runs all experts on
    an intermediate-dim shard, same all-reduce.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops
from ..engine.config import ModelConfig
from ..parallel.state import get_state, tp_all_reduce


class MoEMLP(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        st = get_state()
        tp = st.tp_size
        self.cfg = cfg
        h = cfg.hidden_size
        ie = cfg.moe_intermediate_size or cfg.intermediate_size
        e = cfg.num_experts
        self.top_k = cfg.num_experts_per_tok
        self.num_experts = e
        if tp > 1 and e % tp == 0:
            # expert parallelism: contiguous expert slice per rank
            self.e_local = e // tp
            self.e_base = st.tp_rank * self.e_local
            self.ie_local = ie
        else:
            assert tp == 1 or ie % tp == 0, (ie, tp)
            self.e_local = e
            self.e_base = 0
            self.ie_local = ie // tp
        self.gate = nn.Parameter(torch.empty(e, h, dtype=cfg.dtype),
                                 requires_grad=False)
        # stacked expert weights (local shard):
        # [E_l, 2*IE_l, h] (gate|up) and [E_l, h, IE_l]
        self.w_gate_up = nn.Parameter(
            torch.empty(self.e_local, 2 * self.ie_local, h, dtype=cfg.dtype),
            requires_grad=False)
        self.w_down = nn.Parameter(
            torch.empty(self.e_local, h, self.ie_local, dtype=cfg.dtype),
            requires_grad=False)
        if cfg.moe_bias:   # gpt-oss per-expert biases
            self.b_gate_up = nn.Parameter(
                torch.empty(self.e_local, 2 * self.ie_local, dtype=cfg.dtype),
                requires_grad=False)
            self.b_down = nn.Parameter(
                torch.empty(self.e_local, h, dtype=cfg.dtype),
                requires_grad=False)
        else:
            self.b_gate_up = None
            self.b_down = None
        self.act_mode = 1 if cfg.moe_act == "swiglu_oai" else 0
        # deepseek-v3 noaux_tc routing: per-expert selection bias
        if cfg.moe_routing == "noaux_tc":
            self.e_score_correction_bias = nn.Parameter(
                torch.empty(e, dtype=torch.float32), requires_grad=False)
        # deepseek shared experts: always-on dense SwiGLU of width
        # n_shared * moe_ie, IE-sharded over TP; its down output is a
        # TP-partial like the routed output, so both ride ONE all-reduce
        if cfg.n_shared_experts > 0:
            sie = cfg.n_shared_experts * ie
            assert tp == 1 or sie % tp == 0, (sie, tp)
            self.sie_local = sie // tp
            self.w_shared_gate_up = nn.Parameter(
                torch.empty(2 * self.sie_local, h, dtype=cfg.dtype),
                requires_grad=False)
            self.w_shared_down = nn.Parameter(
                torch.empty(h, self.sie_local, dtype=cfg.dtype),
                requires_grad=False)
        else:
            self.w_shared_gate_up = None

    def _route(self, x: torch.Tensor):
        """Device-only top-k routing + expert sort. Returns
        (sorted_tok int32 [TK], gates f32 [TK], offsets int32 [E+1])."""
        T = x.size(0)
        cfg = self.cfg
        logits = torch.nn.functional.linear(x, self.gate)      # [T, E]
        if cfg.moe_routing == "noaux_tc":
            # deepseek-v3: sigmoid scores; SELECTION uses scores+bias with
            # group-limited top-k; WEIGHTS are the original sigmoid scores
            scores = torch.sigmoid(logits.float())
            sel = scores + self.e_score_correction_bias
            if cfg.n_group > 1:
                gs = sel.view(T, cfg.n_group, -1)
                group_score = gs.topk(min(2, gs.size(-1)), dim=-1)[0].sum(-1)
                gi = group_score.topk(cfg.topk_group, dim=-1)[1]  # [T, kg]
                mask = torch.zeros(T, cfg.n_group, device=x.device,
                                   dtype=torch.bool)
                mask.scatter_(1, gi, True)
                sel = sel.masked_fill(
                    ~mask.unsqueeze(-1).expand_as(gs).reshape(T, -1),
                    float("-inf"))
            topi = sel.topk(self.top_k, dim=-1)[1]             # [T, K]
            topw = scores.gather(1, topi)
            if cfg.moe_norm_topk:
                topw = topw / (topw.sum(-1, keepdim=True) + 1e-20)
            topw = topw * cfg.routed_scaling_factor
        elif cfg.moe_routing == "topk_softmax":
            # gpt-oss: top-k on raw logits, softmax over the selected k
            topl, topi = torch.topk(logits.float(), self.top_k, dim=-1)
            topw = torch.softmax(topl, dim=-1)
        else:
            probs = torch.softmax(logits.float(), dim=-1)
            topw, topi = torch.topk(probs, self.top_k, dim=-1)  # [T, K]
            if cfg.moe_norm_topk:
                topw = topw / topw.sum(-1, keepdim=True)
            topw = topw * cfg.routed_scaling_factor
        flat_e = topi.reshape(-1)                              # [T*K]
        flat_t = torch.arange(T, device=x.device,
                              dtype=torch.int32).repeat_interleave(self.top_k)
        order = torch.argsort(flat_e, stable=True)
        sorted_tok = flat_t[order]
        gates = topw.reshape(-1)[order].float()
        counts = torch.zeros(self.num_experts, device=x.device,
                             dtype=torch.int32)
        counts.scatter_add_(0, flat_e,
                            torch.ones_like(flat_e, dtype=torch.int32))
        offsets = torch.zeros(self.num_experts + 1, device=x.device,
                              dtype=torch.int32)
        torch.cumsum(counts, 0, out=offsets[1:])
        return sorted_tok, gates, offsets

    # fused kernels win at decode-sized batches (weight-streaming
    # bound, graph-capturable); per-expert hipBLASLt GEMMs win at
    # prefill sizes (compute-bound, Tensile ~2x our 2-barrier tile —
    # tools/bench_moe_kernels.py: TK=16384 fused 5.8ms vs loop 3.0ms).
    # Prefill steps run eagerly, so the loop's host-side segment reads
    # are harmless there. KAITO_MOE_FUSED_MAX overrides (high-top-k MoE
    # like deepseek k=6 crosses TK=4096 at decode bs>682, where losing
    # hipGraph capture costs more than the tile-efficiency gap).
    import os as _os
    FUSED_MAX_TOKENS = int(_os.environ.get("KAITO_MOE_FUSED_MAX", "4096"))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T, H = x.shape
        sorted_tok, gates, offsets = self._route(x)
        fused_shapes = x.is_cuda and H % 64 == 0 and self.ie_local % 64 == 0
        # under hipGraph capture the fused kernels are mandatory (the
        # per-expert loop reads segment sizes on the host): deepseek k=6
        # crosses TK=4096 at decode bs>682 and measured 30.9k vs timeout
        # at bs=768 (profiles/r02_perf_notes.md)
        if fused_shapes and (T * self.top_k <= self.FUSED_MAX_TOKENS
                             or torch.cuda.is_current_stream_capturing()):
            out = self._forward_fused(x, sorted_tok, gates, offsets)
        else:
            out = self._forward_loop(x, sorted_tok, gates, offsets)
        if self.w_shared_gate_up is not None:   # deepseek shared experts
            h1 = torch.nn.functional.linear(x, self.w_shared_gate_up)
            out = out + torch.nn.functional.linear(
                ops.silu_and_mul(h1), self.w_shared_down)
        if get_state().tp_size > 1:
            out = tp_all_reduce(out)
        return out

    def _forward_fused(self, x, sorted_tok, gates, offsets):
        """Static-shape HIP grouped GEMMs (graph-capturable)."""
        T, H = x.shape
        TK = T * self.top_k
        # staging reads WHOLE 64-row tiles from unaligned expert offsets:
        # a tile may start at row TK-1 and read through TK+62, so pad a
        # full extra tile beyond the round-up
        pad = (TK + 63) // 64 * 64 + 64
        act = torch.empty(pad, self.ie_local, dtype=x.dtype, device=x.device)
        ops.moe_gate_silu(act, x, self.w_gate_up, sorted_tok, offsets,
                          self.e_base, self.e_local, self.b_gate_up,
                          self.act_mode)
        out32 = torch.zeros(T, H, dtype=torch.float32, device=x.device)
        ops.moe_down_scatter(out32, act, self.w_down, sorted_tok, gates,
                             offsets, self.e_base, self.e_local,
                             self.b_down)
        return out32.to(x.dtype)

    def _forward_loop(self, x, sorted_tok, gates, offsets):
        """Per-expert torch GEMMs (CPU tests / odd shapes). Host sync on
        the segment sizes — never used on the GPU hot path."""
        off = offsets.tolist()
        out = torch.zeros_like(x)
        idx64 = sorted_tok.long()
        for le in range(self.e_local):
            e = self.e_base + le
            s, t = off[e], off[e + 1]
            if s == t:
                continue
            rows = idx64[s:t]
            xs = x[rows]
            h1 = torch.nn.functional.linear(
                xs, self.w_gate_up[le],
                self.b_gate_up[le] if self.b_gate_up is not None else None)
            if self.act_mode == 1:
                g, u = h1.float().chunk(2, dim=-1)
                g = g.clamp(max=7.0)
                u = u.clamp(-7.0, 7.0)
                act = ((u + 1.0) * (g * torch.sigmoid(1.702 * g))).to(x.dtype)
            else:
                act = ops.silu_and_mul(h1.contiguous())
            y = torch.nn.functional.linear(
                act, self.w_down[le],
                self.b_down[le] if self.b_down is not None else None)
            out.index_add_(0, rows,
                           (y.float() * gates[s:t, None]).to(out.dtype))
        return out
