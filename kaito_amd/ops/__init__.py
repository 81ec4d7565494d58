"""kaito_amd.ops — gfx950 HIP kernels with a CPU fp32 reference fallback.

Dispatch policy (engine contract):
  * CUDA/ROCm tensors → the in-tree HIP extension. If the extension is
    missing on a GPU machine this raises loudly — there is NO silent
    eager fallback on GPU.
  * CPU tensors → torch_ref reference implementations (tests / no-GPU dev).
"""
from __future__ import annotations

import torch

from . import torch_ref
from ._build import SO_PATH

_LOADED = False


def extension_available() -> bool:
    return SO_PATH.exists()


def load_extension() -> None:
    """Load (and lazily build) the HIP extension library."""
    global _LOADED
    if _LOADED:
        return
    if not SO_PATH.exists():
        from ._build import build
        build()
    torch.ops.load_library(str(SO_PATH))
    _LOADED = True


def _require_ext() -> None:
    if not _LOADED:
        if not torch.cuda.is_available():
            raise RuntimeError(
                "kaito_amd.ops: got a CUDA tensor but torch.cuda is unavailable")
        load_extension()


# ---------------------------------------------------------------- public ops

def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        _require_ext()
        out = torch.empty_like(x)
        torch.ops.kaito.rms_norm(out, x, weight, eps)
        return out
    return torch_ref.rms_norm(x, weight, eps)


def fused_add_rms_norm(x: torch.Tensor, residual: torch.Tensor,
                       weight: torch.Tensor, eps: float):
    """Returns (normed, residual). On GPU, `residual` is updated IN PLACE
    (residual += x) and returned."""
    if x.is_cuda:
        _require_ext()
        out = torch.empty_like(x)
        torch.ops.kaito.fused_add_rms_norm(out, x, residual, weight, eps)
        return out, residual
    return torch_ref.fused_add_rms_norm(x, residual, weight, eps)


def rotary_embedding(positions: torch.Tensor, q: torch.Tensor, k: torch.Tensor,
                     head_dim: int, cos_sin_cache: torch.Tensor):
    """In-place on GPU; returns (q, k)."""
    if q.is_cuda:
        _require_ext()
        torch.ops.kaito.rotary_embedding(positions, q, k, head_dim, cos_sin_cache)
        return q, k
    return torch_ref.rotary_embedding(positions, q, k, head_dim, cos_sin_cache)


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        _require_ext()
        d = x.size(-1) // 2
        out = torch.empty(*x.shape[:-1], d, dtype=x.dtype, device=x.device)
        torch.ops.kaito.silu_and_mul(out, x)
        return out
    return torch_ref.silu_and_mul(x)


def reshape_and_cache(k: torch.Tensor, v: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, slot_mapping: torch.Tensor) -> None:
    """k/v: [T, KH, D] (rows may be strided views into a fused qkv buffer)."""
    if k.is_cuda:
        _require_ext()
        torch.ops.kaito.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)
        return
    torch_ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


# decode-attention implementation: "sp" = split-phase (r02, K-stream →
# exp pass → V-stream, no softmax chain in the memory loops), "fused" =
# the r01 single-pass online-softmax kernel. Overridable for A/B runs.
import os as _os
_PA_IMPL = _os.environ.get("KAITO_PA_IMPL", "sp")
# fused W4A16 GEMM for M>4: measured slower than dequant+hipBLASLt on
# MI355X (tools/bench_w4a16.py) — off unless explicitly enabled
W4A16_FUSED_GEMM = _os.environ.get("KAITO_W4A16_FUSED", "0") == "1"


def _sinks_arg(sinks, device):
    if sinks is None:
        return torch.empty(0, dtype=torch.float32, device=device)
    return sinks


def paged_attention(q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
                    block_tables: torch.Tensor, seq_lens: torch.Tensor,
                    scale: float, window: int = 0,
                    sinks: torch.Tensor | None = None) -> torch.Tensor:
    if q.is_cuda:
        _require_ext()
        out = torch.empty_like(q)
        if _PA_IMPL == "sp" or window > 0 or sinks is not None \
                or k_cache.dtype == torch.uint8:
            # window/sink support lives in the split-phase kernel only
            torch.ops.kaito.paged_attention_sp(
                out, q, k_cache, v_cache, block_tables, seq_lens, scale,
                window, _sinks_arg(sinks, q.device))
        else:
            torch.ops.kaito.paged_attention(out, q, k_cache, v_cache,
                                            block_tables, seq_lens, scale)
        return out
    return torch_ref.paged_attention(q, k_cache, v_cache, block_tables,
                                     seq_lens, scale, window, sinks)


def mla_decode(q: torch.Tensor, cache: torch.Tensor,
               block_tables: torch.Tensor, seq_lens: torch.Tensor,
               scale: float, r: int) -> torch.Tensor:
    """Absorbed MLA decode over the compressed latent cache (DeepSeek).
    q: [T, H, r+rope]; cache: [NB, BS, r+rope] bf16. Returns [T, H, r]."""
    if q.is_cuda:
        _require_ext()
        out = torch.empty(q.size(0), q.size(1), r, dtype=q.dtype,
                          device=q.device)
        torch.ops.kaito.mla_decode(out, q, cache, block_tables, seq_lens,
                                   scale)
        return out
    return torch_ref.mla_decode(q, cache, block_tables, seq_lens, scale, r)


def mla_cache_write(cache: torch.Tensor, c_kv: torch.Tensor,
                    k_pe: torch.Tensor, slots: torch.Tensor) -> None:
    """Scatter latent rows (c_kv ‖ k_pe) into the paged MLA cache;
    negative slots (graph-bucket padding) are skipped."""
    _require_ext()
    torch.ops.kaito.mla_cache_write(cache, c_kv, k_pe, slots)


def prefill_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      cu_seqlens: torch.Tensor, scale: float,
                      max_seqlen: int | None = None, window: int = 0,
                      sinks: torch.Tensor | None = None) -> torch.Tensor:
    if q.is_cuda:
        _require_ext()
        out = torch.empty_like(q)
        tile_seq, tile_qbase = _build_tiles(cu_seqlens)
        torch.ops.kaito.prefill_attention(out, q, k, v, tile_seq, tile_qbase,
                                          cu_seqlens, scale, window,
                                          _sinks_arg(sinks, q.device))
        return out
    return torch_ref.prefill_attention(q, k, v, cu_seqlens, scale, window,
                                       sinks)


def context_attention(q: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, cu_seqlens_q: torch.Tensor,
                      kv_lens: torch.Tensor, block_tables: torch.Tensor,
                      scale: float, window: int = 0,
                      sinks: torch.Tensor | None = None) -> torch.Tensor:
    """Suffix-query causal attention over the paged cache (chunked
    prefill / prefix-cache continuation)."""
    if q.is_cuda:
        _require_ext()
        out = torch.empty_like(q)
        tile_seq, tile_qbase = _build_tiles(cu_seqlens_q)
        torch.ops.kaito.context_attention(out, q, k_cache, v_cache, tile_seq,
                                          tile_qbase, cu_seqlens_q, kv_lens,
                                          block_tables, scale, window,
                                          _sinks_arg(sinks, q.device))
        return out
    return torch_ref.context_attention(q, k_cache, v_cache, cu_seqlens_q,
                                       kv_lens, block_tables, scale, window,
                                       sinks)


def gelu_and_mul(x: torch.Tensor) -> torch.Tensor:
    """GeGLU: gelu_tanh(x[..., :d]) * x[..., d:] (gemma MLPs)."""
    if x.is_cuda:
        _require_ext()
        d = x.size(-1) // 2
        out = torch.empty(*x.shape[:-1], d, dtype=x.dtype, device=x.device)
        torch.ops.kaito.gelu_and_mul(out, x)
        return out
    return torch_ref.gelu_and_mul(x)


def gelu(x: torch.Tensor) -> torch.Tensor:
    """Plain tanh-approx GELU (phi-2 ungated MLP)."""
    if x.is_cuda:
        _require_ext()
        out = torch.empty_like(x)
        torch.ops.kaito.gelu(out, x.contiguous())
        return out
    return torch_ref.gelu_tanh(x)


def layer_norm(x: torch.Tensor, weight: torch.Tensor,
               bias: torch.Tensor | None, eps: float) -> torch.Tensor:
    if x.is_cuda:
        _require_ext()
        out = torch.empty_like(x)
        b = bias if bias is not None else \
            torch.empty(0, dtype=x.dtype, device=x.device)
        torch.ops.kaito.layer_norm(out, x, weight, b, eps)
        return out
    return torch_ref.layer_norm(x, weight, bias, eps)


def fused_add_layer_norm(x: torch.Tensor, residual: torch.Tensor,
                         weight: torch.Tensor, bias: torch.Tensor | None,
                         eps: float):
    """Returns (normed, residual); residual += x in place on GPU."""
    if x.is_cuda:
        _require_ext()
        out = torch.empty_like(x)
        b = bias if bias is not None else \
            torch.empty(0, dtype=x.dtype, device=x.device)
        torch.ops.kaito.fused_add_layer_norm(out, x, residual, weight, b, eps)
        return out, residual
    new_res = (x.float() + residual.float()).to(x.dtype)
    return torch_ref.layer_norm(new_res, weight, bias, eps), new_res


def mfma_tile_gemm(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    _require_ext()
    return torch.ops.kaito.mfma_tile_gemm(a, b)


QTILE = 64  # must match prefill_attention.hip


def _build_tiles(cu_seqlens: torch.Tensor):
    """Map varlen batch → list of 64-row Q tiles: (seq_idx, q_base)."""
    cs = cu_seqlens.tolist()
    seqs, bases = [], []
    for i in range(len(cs) - 1):
        L = cs[i + 1] - cs[i]
        for q0 in range(0, L, QTILE):
            seqs.append(i)
            bases.append(q0)
    dev = cu_seqlens.device
    return (torch.tensor(seqs, dtype=torch.int32, device=dev),
            torch.tensor(bases, dtype=torch.int32, device=dev))


def moe_gate_silu(act: torch.Tensor, x: torch.Tensor,
                  w_gate_up: torch.Tensor, sorted_ids: torch.Tensor,
                  offsets: torch.Tensor, e_base: int,
                  n_local_experts: int, bias: torch.Tensor | None = None,
                  act_mode: int = 0) -> None:
    """Fused MoE stage 1 (GPU only): gather rows by sorted token ids,
    gate/up grouped GEMM per expert, activation epilogue into `act`
    (act_mode 0 = SwiGLU, 1 = gpt-oss clamped swiglu; optional
    per-expert [E, 2*IE] bias)."""
    _require_ext()
    b = bias if bias is not None else         torch.empty(0, dtype=x.dtype, device=x.device)
    torch.ops.kaito.moe_gate_silu(act, x, w_gate_up, sorted_ids, offsets,
                                  b, act_mode, e_base, n_local_experts)


def moe_down_scatter(out: torch.Tensor, act: torch.Tensor,
                     w_down: torch.Tensor, sorted_ids: torch.Tensor,
                     gates: torch.Tensor, offsets: torch.Tensor,
                     e_base: int, n_local_experts: int,
                     bias: torch.Tensor | None = None) -> None:
    """Fused MoE stage 2 (GPU only): down grouped GEMM (+ optional
    [E, H] bias) + gated f32 atomic scatter into `out`."""
    _require_ext()
    b = bias if bias is not None else         torch.empty(0, dtype=act.dtype, device=act.device)
    torch.ops.kaito.moe_down_scatter(out, act, w_down, sorted_ids, gates,
                                     offsets, b, e_base, n_local_experts)


def w4a16_gemv(x: torch.Tensor, qweight: torch.Tensor, scales: torch.Tensor,
               zeros: torch.Tensor, group: int) -> torch.Tensor:
    """Group-quantized W4A16 linear, small M (decode): out = x @ W^T with
    W dequantized on the fly inside the kernel."""
    if x.is_cuda:
        _require_ext()
        out = torch.empty(x.size(0), qweight.size(0), dtype=x.dtype,
                          device=x.device)
        torch.ops.kaito.w4a16_gemv(out, x, qweight, scales, zeros, group)
        return out
    return torch_ref.w4a16_gemv(x, qweight, scales, zeros, group)


def w4a16_dequant(qweight: torch.Tensor, scales: torch.Tensor,
                  zeros: torch.Tensor, group: int,
                  out: torch.Tensor = None) -> torch.Tensor:
    """Dequantize packed 4-bit weights to bf16 [N, K] (feeds hipBLASLt
    MFMA for large-M GEMMs)."""
    if qweight.is_cuda:
        _require_ext()
        if out is None:
            out = torch.empty(qweight.size(0), qweight.size(1) * 8,
                              dtype=torch.bfloat16, device=qweight.device)
        torch.ops.kaito.w4a16_dequant(out, qweight, scales, zeros, group)
        return out
    ref = torch_ref.w4a16_unpack(qweight, scales, zeros, group)
    return ref.to(torch.bfloat16) if out is None else out.copy_(ref)


def w4a16_gemm(x: torch.Tensor, qweight: torch.Tensor, scales: torch.Tensor,
               zeros: torch.Tensor, group: int) -> torch.Tensor:
    """Fused inline-dequant MFMA GEMM (GPU): y = x @ dequant(W)^T.
    The 4-bit weights never round-trip through a full-precision scratch
    buffer. CPU falls back to the unpack reference."""
    if x.is_cuda:
        _require_ext()
        out = torch.empty(x.size(0), qweight.size(0), dtype=x.dtype,
                          device=x.device)
        torch.ops.kaito.w4a16_gemm(out, x.contiguous(), qweight, scales,
                                   zeros, group)
        return out
    return torch_ref.w4a16_gemv(x, qweight, scales, zeros, group)
