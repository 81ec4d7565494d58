"""W4A16 group quantization: pack/repack + the QuantLinear module.

Covers the reference catalog's AWQ/GPTQ presets (SURVEY.md §2.3
"quantized GEMM" row; supported_models.yaml qwen3-8b-awq etc.). Native
layout is kernel-first (ops/csrc/w4a16.hip): qweight u32 [N, K/8] with 8
consecutive K nibbles per word, fp32 scales/zeros [N, K/G], w = s*q - z.
AWQ checkpoints repack into it at load time.
"""
from __future__ import annotations

from typing import Tuple

import torch

from .. import ops

# Public AWQ packing (llm-awq/AutoAWQ): nibble position i of each i32 word
# holds LOGICAL column AWQ_ORDER[i]. Unpacking to logical column order
# therefore reads nibble positions in the INVERSE permutation,
# AWQ_REVERSE_ORDER (logical col c lives at nibble AWQ_REVERSE_ORDER[c]) —
# the same table vLLM uses. The permutation has order 3, so using the
# forward table for unpacking scrambles weights within each 8-column group.
AWQ_ORDER = (0, 2, 4, 6, 1, 3, 5, 7)
AWQ_REVERSE_ORDER = (0, 4, 1, 5, 2, 6, 3, 7)


def quantize_w4(weight: torch.Tensor, group: int = 128
                ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Asymmetric 4-bit group quantization of [N, K] → native layout.
    Returns (qweight u32 [N, K/8], scales f32 [N, K/G], zeros f32)."""
    N, K = weight.shape
    assert K % group == 0 and group % 8 == 0
    w = weight.float().reshape(N, K // group, group)
    wmax = w.amax(dim=-1)
    wmin = w.amin(dim=-1)
    scale = (wmax - wmin).clamp(min=1e-8) / 15.0
    zq = (-wmin / scale).round().clamp(0, 15)
    q = (w / scale.unsqueeze(-1) + zq.unsqueeze(-1)).round().clamp(0, 15)
    q = q.reshape(N, K).to(torch.int64)
    shifts = torch.arange(8, dtype=torch.int64, device=q.device) * 4
    packed = (q.reshape(N, K // 8, 8) << shifts).sum(dim=-1)
    qweight = packed.to(torch.int32)          # bit pattern == uint32
    zeros = (scale * zq).float()              # z folded: w = s*q - z
    return qweight, scale.float(), zeros


def repack_awq(qweight_awq: torch.Tensor, qzeros_awq: torch.Tensor,
               scales_awq: torch.Tensor, group: int = 128
               ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Repack a public-AWQ checkpoint (qweight i32 [K, N/8] nibbles along N
    in AWQ_ORDER; qzeros i32 [K/G, N/8]; scales [K/G, N]) into the native
    layout."""
    K, nw = qweight_awq.shape
    N = nw * 8

    def unpack_n(t: torch.Tensor) -> torch.Tensor:
        # [R, N/8] i32 → [R, N] int in logical n order: logical column c
        # of each group is stored at nibble position AWQ_REVERSE_ORDER[c]
        cols = []
        for j in AWQ_REVERSE_ORDER:
            cols.append((t >> (4 * j)) & 0xF)
        return torch.stack(cols, dim=-1).reshape(t.shape[0], N)

    q = unpack_n(qweight_awq.long())            # [K, N]
    zq = unpack_n(qzeros_awq.long())            # [K/G, N]
    s = scales_awq.float()                      # [K/G, N]
    qn = q.T.contiguous()                       # [N, K]
    shifts = torch.arange(8, dtype=torch.int64, device=qn.device) * 4
    packed = (qn.reshape(N, K // 8, 8) << shifts).sum(dim=-1).to(torch.int32)
    scales = s.T.contiguous()                   # [N, K/G]
    zeros = (scales * zq.T.float())
    return packed, scales, zeros


class QuantLinear(torch.nn.Module):
    """W4A16 linear: HIP GEMV for decode-sized M, dequant + MFMA GEMM
    (hipBLASLt) beyond. Drop-in for a bias-free nn.Linear."""

    # measured crossover on MI355X (profiles/r01_decode_profile.md):
    # GEMV wins to M=4 (10-24us vs ~38us dequant+GEMM); beyond, tile
    # dequant into the MFMA GEMM path wins
    GEMV_MAX_M = 4

    def __init__(self, qweight, scales, zeros, group: int = 128):
        super().__init__()
        self.register_buffer("qweight", qweight)
        self.register_buffer("scales", scales)
        self.register_buffer("zeros", zeros)
        self.group = group
        self.out_features = qweight.size(0)
        self.in_features = qweight.size(1) * 8

    @classmethod
    def from_float(cls, weight: torch.Tensor, group: int = 128):
        return cls(*quantize_w4(weight, group), group)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shape = x.shape[:-1]
        x2 = x.reshape(-1, self.in_features)
        if x2.size(0) <= self.GEMV_MAX_M:
            y = ops.w4a16_gemv(x2, self.qweight, self.scales, self.zeros,
                               self.group)
        elif x.is_cuda and self.out_features % 64 == 0 \
                and self.in_features % 64 == 0 and ops.W4A16_FUSED_GEMM:
            # fused inline-dequant MFMA GEMM — measured SLOWER than
            # dequant+hipBLASLt at M>=64 (Tensile ~1.1 PF vs this 64^2
            # tile's 403 TF, tools/bench_w4a16.py); kept env-gated
            # (KAITO_W4A16_FUSED=1) for further schedule work
            y = ops.w4a16_gemm(x2, self.qweight, self.scales, self.zeros,
                               self.group)
        else:
            w = ops.w4a16_dequant(self.qweight, self.scales, self.zeros,
                                  self.group)
            y = torch.nn.functional.linear(x2.to(w.dtype), w)
        return y.reshape(*shape, self.out_features).to(x.dtype)


def quantize_parallel_linears(model: torch.nn.Module, group: int = 128,
                              suffixes=("qkv_proj", "o_proj",
                                        "gate_up_proj", "down_proj")) -> int:
    """Quantize the engine's TP-parallel linears in place (weight-only W4;
    PTQ of whatever weights are loaded). Returns modules converted."""
    n = 0
    for name, mod in model.named_modules():
        if any(name.endswith(sfx) for sfx in suffixes) and \
                hasattr(mod, "quantize_") and \
                mod.weight.numel() and mod.weight.size(1) % group == 0:
            mod.quantize_(group)
            n += 1
    return n


def quantize_model_linears(model: torch.nn.Module, group: int = 128,
                           suffixes=("qkv_proj", "o_proj", "gate_up_proj",
                                     "down_proj")) -> int:
    """Swap matching nn.Linear weights for QuantLinear (weight-only W4).
    Returns the number of modules converted."""
    n = 0
    for parent in model.modules():
        for name, child in list(parent.named_children()):
            if any(name.endswith(sfx) for sfx in suffixes) and \
                    isinstance(child, torch.nn.Linear) and \
                    child.in_features % group == 0:
                ql = QuantLinear.from_float(child.weight.data, group)
                ql = ql.to(child.weight.device)
                setattr(parent, name, ql)
                n += 1
    return n


def load_awq_checkpoint(model: torch.nn.Module, path: str) -> int:
    """Load a public-AWQ safetensors checkpoint into the engine's
    (fused, TP-sharded) quantized linears. AWQ stores per-module
    qweight i32 [K, N/8] / qzeros i32 [K/G, N/8] / scales [K/G, N]
    (nibbles along N in AWQ_ORDER); we repack each projection into the
    native layout, fuse q/k/v and gate/up along N, and shard for TP
    (column-parallel along N rows, row-parallel along packed-K words).
    Returns the number of modules loaded."""
    import json
    import os
    from pathlib import Path
    from safetensors.torch import safe_open
    from ..parallel.state import get_state

    with open(os.path.join(path, "config.json")) as f:
        hf = json.load(f)
    qc = hf.get("quantization_config", {})
    group = qc.get("group_size", 128)
    tensors = {}
    for f in sorted(Path(path).glob("*.safetensors")):
        with safe_open(str(f), framework="pt", device="cpu") as sf:
            for k in sf.keys():
                tensors[k] = sf.get_tensor(k)

    def native(prefix):
        def get(sfx):
            for p in (f"model.{prefix}.{sfx}", f"{prefix}.{sfx}"):
                if p in tensors:
                    return tensors[p]
            raise KeyError(f"{prefix}.{sfx}")
        return repack_awq(get("qweight"), get("qzeros"),
                          get("scales"), group)

    st = get_state()
    tp, rank = st.tp_size, st.tp_rank

    def col_shard(t):                      # [N, ...] → rank's N rows
        n = t.size(0) // tp
        return t[rank * n:(rank + 1) * n].contiguous()

    def row_shard(qw, sc, z):              # along K: packed words + groups
        kw = qw.size(1) // tp
        kg = sc.size(1) // tp
        return (qw[:, rank * kw:(rank + 1) * kw].contiguous(),
                sc[:, rank * kg:(rank + 1) * kg].contiguous(),
                z[:, rank * kg:(rank + 1) * kg].contiguous())

    n_loaded = 0
    for name, mod in model.named_modules():
        if not hasattr(mod, "quantize_from_packed"):
            continue
        lid = name.split(".")[1] if name.startswith("layers.") else None
        if name.endswith("qkv_proj"):
            parts = [native(f"layers.{lid}.self_attn.{p}")
                     for p in ("q_proj", "k_proj", "v_proj")]
            qw = torch.cat([col_shard(p[0]) for p in parts])
            sc = torch.cat([col_shard(p[1]) for p in parts])
            z = torch.cat([col_shard(p[2]) for p in parts])
        elif name.endswith("gate_up_proj"):
            parts = [native(f"layers.{lid}.mlp.{p}")
                     for p in ("gate_proj", "up_proj")]
            qw = torch.cat([col_shard(p[0]) for p in parts])
            sc = torch.cat([col_shard(p[1]) for p in parts])
            z = torch.cat([col_shard(p[2]) for p in parts])
        elif name.endswith("o_proj"):
            qw, sc, z = row_shard(*native(f"layers.{lid}.self_attn.o_proj"))
        elif name.endswith("down_proj"):
            qw, sc, z = row_shard(*native(f"layers.{lid}.mlp.down_proj"))
        else:
            continue
        mod.quantize_from_packed(qw, sc, z, group)
        n_loaded += 1
    return n_loaded
