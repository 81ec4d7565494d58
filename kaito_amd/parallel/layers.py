"""Tensor-parallel linear layers (RCCL over xGMI).

Column-parallel: weight split along output dim, no comm on forward.
Row-parallel: weight split along input dim, all-reduce on forward.
At tp=1 these are plain GEMMs (hipBLASLt via torch.nn.functional.linear).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .state import get_state, tp_all_reduce
from ..engine import lora as lora_mod


class _QuantMixin:
    """Weight-only W4A16 for the parallel linears: quantize_() packs the
    rank-local weight shard into the native 4-bit layout and switches the
    matmul to the HIP GEMV (decode M) / dequant+MFMA (large M) path. TP
    comms, bias, and the LoRA hook are untouched — quantization is purely
    the local GEMM."""

    _quantized = False

    def quantize_(self, group: int = 128) -> None:
        from ..models.quant import quantize_w4
        w = self.weight.data
        assert w.size(1) % group == 0, (w.shape, group)
        qw, sc, z = quantize_w4(w.float(), group)
        self.quantize_from_packed(qw, sc, z, group)

    def quantize_from_packed(self, qweight, scales, zeros,
                             group: int = 128) -> None:
        """Install pre-packed 4-bit weights (native layout) — the AWQ
        checkpoint load path."""
        from ..models.quant import QuantLinear
        dev = self.weight.device
        dtype = self.weight.dtype
        self.register_buffer("qweight", qweight.to(dev))
        self.register_buffer("scales", scales.to(dev))
        self.register_buffer("zeros", zeros.to(dev))
        self.q_group = group
        self.weight = nn.Parameter(torch.empty(0, dtype=dtype, device=dev),
                                   requires_grad=False)  # drop bf16 copy
        self._quantized = True
        self._gemv_max_m = QuantLinear.GEMV_MAX_M

    def _qmatmul(self, x: torch.Tensor) -> torch.Tensor:
        from .. import ops
        shape = x.shape[:-1]
        x2 = x.reshape(-1, x.shape[-1])
        if x2.size(0) <= self._gemv_max_m:
            y = ops.w4a16_gemv(x2.contiguous(), self.qweight, self.scales,
                               self.zeros, self.q_group)
        elif x.is_cuda and self.qweight.size(0) % 64 == 0 \
                and x2.shape[-1] % 64 == 0 and ops.W4A16_FUSED_GEMM:
            # env-gated: see models/quant.py (dequant+hipBLASLt measured
            # faster at mid/large M)
            y = ops.w4a16_gemm(x2.contiguous(), self.qweight, self.scales,
                               self.zeros, self.q_group)
        else:
            w = ops.w4a16_dequant(self.qweight, self.scales, self.zeros,
                                  self.q_group)
            y = F.linear(x2.to(w.dtype), w)
        return y.reshape(*shape, -1).to(x.dtype)


class ColumnParallelLinear(_QuantMixin, nn.Module):
    def __init__(self, in_features: int, out_features: int, bias: bool = False,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        tp = get_state().tp_size
        assert out_features % tp == 0, (out_features, tp)
        self.in_features = in_features
        self.out_features_per_rank = out_features // tp
        self.weight = nn.Parameter(
            torch.empty(self.out_features_per_rank, in_features, dtype=dtype),
            requires_grad=False)
        self.bias = nn.Parameter(
            torch.empty(self.out_features_per_rank, dtype=dtype),
            requires_grad=False) if bias else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._quantized:
            out = self._qmatmul(x)
            if self.bias is not None:
                out = out + self.bias
        else:
            out = F.linear(x, self.weight, self.bias)
        return lora_mod.maybe_apply(self, x, out)


class RowParallelLinear(_QuantMixin, nn.Module):
    # set True by the model builder on linears whose output feeds a
    # fused allreduce+RMSNorm call site (attention o_proj, MLP
    # down_proj): when a one-shot group is active and the batch fits its
    # staging window, forward returns the LOCAL partial and the fused
    # kernel performs the reduction (parallel/one_shot.py).
    fuse_norm = False

    def __init__(self, in_features: int, out_features: int, bias: bool = False,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        tp = get_state().tp_size
        assert in_features % tp == 0, (in_features, tp)
        self.in_features_per_rank = in_features // tp
        self.out_features = out_features
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_features_per_rank, dtype=dtype),
            requires_grad=False)
        # bias added once (after reduce) on rank 0's shard only
        self.bias = nn.Parameter(
            torch.empty(out_features, dtype=dtype),
            requires_grad=False) if bias else None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self._qmatmul(x) if self._quantized else F.linear(x, self.weight)
        out = lora_mod.maybe_apply(self, x, out)
        if self.fuse_norm and self.bias is None and get_state().tp_size > 1:
            from . import one_shot
            if one_shot.defer(out.reshape(-1, out.shape[-1]).shape[0]):
                # deferred: the following fused one-shot call reduces
                return out
        out = tp_all_reduce(out)
        if self.bias is not None:
            out = out + self.bias
        return out


class VocabParallelEmbedding(nn.Module):
    """Embedding split along vocab; out-of-shard ids contribute 0, summed by
    all-reduce."""

    def __init__(self, num_embeddings: int, embedding_dim: int,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        st = get_state()
        tp = st.tp_size
        assert num_embeddings % tp == 0
        self.per_rank = num_embeddings // tp
        self.start = st.tp_rank * self.per_rank
        self.weight = nn.Parameter(
            torch.empty(self.per_rank, embedding_dim, dtype=dtype),
            requires_grad=False)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        if get_state().tp_size == 1:
            return F.embedding(ids, self.weight)
        local = ids - self.start
        mask = (local < 0) | (local >= self.per_rank)
        local = local.clamp(0, self.per_rank - 1)
        out = F.embedding(local, self.weight)
        out = out.masked_fill(mask.unsqueeze(-1), 0)
        return tp_all_reduce(out)
