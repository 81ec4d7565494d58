"""Mixture-of-experts FFN (Mixtral / Qwen-MoE / gpt-oss class).

Replaces vLLM's fused-MoE CUDA path for the MoE presets in the reference
catalog (SURVEY.md §2.3 note: gpt-oss-120b / Qwen MoE presets in
supported_models.yaml). Round-1 implementation: top-k softmax gating +
expert-sorted grouped GEMMs (torch/hipBLASLt); tokens are sorted by expert
so each expert runs one contiguous GEMM (the standard moe-align layout a
future fused HIP kernel will consume directly). EP/TP sharding of experts
lands with the fused kernel.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops
from ..engine.config import ModelConfig
from ..parallel.state import get_state


class MoEMLP(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        assert get_state().tp_size == 1, \
            "MoE presets currently require tp=1 (expert parallelism lands " \
            "with the fused HIP MoE kernel)"
        self.cfg = cfg
        h = cfg.hidden_size
        ie = cfg.moe_intermediate_size or cfg.intermediate_size
        e = cfg.num_experts
        self.top_k = cfg.num_experts_per_tok
        self.gate = nn.Parameter(torch.empty(e, h, dtype=cfg.dtype),
                                 requires_grad=False)
        # stacked expert weights: [E, 2*ie, h] (gate|up) and [E, h, ie]
        self.w_gate_up = nn.Parameter(
            torch.empty(e, 2 * ie, h, dtype=cfg.dtype), requires_grad=False)
        self.w_down = nn.Parameter(
            torch.empty(e, h, ie, dtype=cfg.dtype), requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T, H = x.shape
        logits = torch.nn.functional.linear(x, self.gate)      # [T, E]
        probs = torch.softmax(logits.float(), dim=-1)
        topw, topi = torch.topk(probs, self.top_k, dim=-1)     # [T, K]
        topw = (topw / topw.sum(-1, keepdim=True)).to(x.dtype)

        # moe-align: flatten (token, k) pairs, sort by expert
        flat_e = topi.reshape(-1)                              # [T*K]
        flat_t = torch.arange(T, device=x.device).repeat_interleave(self.top_k)
        order = torch.argsort(flat_e, stable=True)
        se, st_idx = flat_e[order], flat_t[order]
        counts = torch.bincount(se, minlength=self.cfg.num_experts)

        xs = x[st_idx]                                         # [T*K, H]
        out_sorted = torch.empty_like(xs)
        start = 0
        for eid, n in enumerate(counts.tolist()):
            if n == 0:
                continue
            sl = slice(start, start + n)
            h1 = torch.nn.functional.linear(xs[sl], self.w_gate_up[eid])
            act = ops.silu_and_mul(h1.contiguous())
            out_sorted[sl] = torch.nn.functional.linear(act, self.w_down[eid])
            start += n

        # scatter-add back with gating weights
        w_sorted = topw.reshape(-1)[order].unsqueeze(1)
        out = torch.zeros_like(x)
        out.index_add_(0, st_idx, out_sorted * w_sorted)
        return out
