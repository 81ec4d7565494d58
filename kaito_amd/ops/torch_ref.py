"""Pure-PyTorch fp32 reference implementations of every HIP op.

Used (a) on CPU so the engine/scheduler stack is testable without a GPU and
(b) as the numerics oracle for the GPU kernels (tests compare HIP bf16
kernels against these fp32 references).
"""
from __future__ import annotations

import torch


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * weight.float()
    return out.to(x.dtype)


def fused_add_rms_norm(x: torch.Tensor, residual: torch.Tensor,
                       weight: torch.Tensor, eps: float):
    """Returns (normed, new_residual). Residual update happens in bf16 to
    match the kernel's residual-stream precision."""
    new_res = (x.float() + residual.float()).to(x.dtype)
    return rms_norm(new_res, weight, eps), new_res


def rotary_embedding(positions: torch.Tensor, q: torch.Tensor, k: torch.Tensor,
                     head_dim: int, cos_sin_cache: torch.Tensor):
    """Neox/llama rotate-half RoPE. q: [T, QH*D] or [T, QH, D]; in-place-like
    (returns rotated copies). cos_sin_cache: [max_pos, R] = [cos | sin]."""
    rot = cos_sin_cache.size(1)
    half = rot // 2
    cs = cos_sin_cache[positions]          # [T, R]
    cos = cs[:, :half].float()             # [T, half]
    sin = cs[:, half:].float()

    def _apply(t: torch.Tensor) -> torch.Tensor:
        shp = t.shape
        x = t.reshape(shp[0], -1, head_dim).float()
        x1 = x[..., :half]
        x2 = x[..., half:rot]
        c = cos.unsqueeze(1)
        s = sin.unsqueeze(1)
        o1 = x1 * c - x2 * s
        o2 = x2 * c + x1 * s
        out = torch.cat([o1, o2, x[..., rot:]], dim=-1)
        return out.reshape(shp).to(t.dtype)

    return _apply(q), _apply(k)


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    d = x.size(-1) // 2
    gate, up = x[..., :d].float(), x[..., d:].float()
    return (torch.nn.functional.silu(gate) * up).to(x.dtype)


def gelu_and_mul(x: torch.Tensor) -> torch.Tensor:
    """GeGLU (gemma): gelu_tanh(gate) * up."""
    d = x.size(-1) // 2
    gate, up = x[..., :d].float(), x[..., d:].float()
    return (torch.nn.functional.gelu(gate, approximate="tanh") * up
            ).to(x.dtype)


def gelu_tanh(x: torch.Tensor) -> torch.Tensor:
    """Plain tanh-approx GELU (phi-2's ungated MLP activation)."""
    return torch.nn.functional.gelu(x.float(), approximate="tanh").to(x.dtype)


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias, eps: float
               ) -> torch.Tensor:
    """LayerNorm with optional bias (phi-2/falcon norm layers)."""
    xf = x.float()
    mu = xf.mean(-1, keepdim=True)
    var = (xf - mu).pow(2).mean(-1, keepdim=True)
    out = (xf - mu) * torch.rsqrt(var + eps) * weight.float()
    if bias is not None:
        out = out + bias.float()
    return out.to(x.dtype)


def reshape_and_cache(k: torch.Tensor, v: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, slot_mapping: torch.Tensor):
    """k/v: [T, KH, D]; caches: [B, KH, BS, D]."""
    bs = k_cache.size(2)
    mask = slot_mapping >= 0
    slots = slot_mapping[mask]
    blocks = torch.div(slots, bs, rounding_mode="floor")
    offs = slots % bs
    k_cache[blocks, :, offs] = k[mask].to(k_cache.dtype)
    v_cache[blocks, :, offs] = v[mask].to(v_cache.dtype)


def paged_attention(q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
                    block_tables: torch.Tensor, seq_lens: torch.Tensor,
                    scale: float, window: int = 0,
                    sinks=None) -> torch.Tensor:
    """q: [T, QH, D] (one token/seq). Returns [T, QH, D].
    window>0: sliding-window attention — only the last `window` tokens
    are attended. sinks: [QH] learned attention-sink logits (gpt-oss)
    folded into the softmax denominator."""
    T, QH, D = q.shape
    KH = k_cache.size(1)
    BS = k_cache.size(2)
    G = QH // KH
    out = torch.empty_like(q)
    for i in range(T):
        L = int(seq_lens[i])
        start = max(0, L - window) if window > 0 else 0
        nb = (L + BS - 1) // BS
        blocks = block_tables[i, :nb].long()
        keys = k_cache[blocks].permute(1, 0, 2, 3).reshape(KH, nb * BS, D)[:, start:L]
        vals = v_cache[blocks].permute(1, 0, 2, 3).reshape(KH, nb * BS, D)[:, start:L]
        qh = q[i].float()                                # [QH, D]
        kx = keys.float().repeat_interleave(G, dim=0)    # [QH, Lw, D]
        vx = vals.float().repeat_interleave(G, dim=0)
        s = torch.einsum("hd,hld->hl", qh, kx) * scale
        if sinks is not None:
            s = torch.cat([sinks.float().unsqueeze(1), s], dim=1)
            p = torch.softmax(s, dim=-1)[:, 1:]          # sink absorbs mass
        else:
            p = torch.softmax(s, dim=-1)
        out[i] = torch.einsum("hl,hld->hd", p, vx).to(q.dtype)
    return out


def prefill_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      cu_seqlens: torch.Tensor, scale: float,
                      window: int = 0, sinks=None) -> torch.Tensor:
    """Varlen causal attention. q: [T, QH, D], k/v: [T, KH, D].
    window>0: sliding window (k in (qpos-window, qpos]); sinks: [QH]."""
    T, QH, D = q.shape
    KH = k.size(1)
    G = QH // KH
    out = torch.empty_like(q)
    cs = cu_seqlens.tolist()
    for b in range(len(cs) - 1):
        s0, s1 = cs[b], cs[b + 1]
        L = s1 - s0
        qs = q[s0:s1].float().transpose(0, 1)                      # [QH, L, D]
        ks = k[s0:s1].float().transpose(0, 1).repeat_interleave(G, 0)
        vs = v[s0:s1].float().transpose(0, 1).repeat_interleave(G, 0)
        att = torch.einsum("hqd,hkd->hqk", qs, ks) * scale
        qpos = torch.arange(L, device=q.device).unsqueeze(1)
        kpos = torch.arange(L, device=q.device).unsqueeze(0)
        mask = kpos > qpos
        if window > 0:
            mask = mask | (kpos <= qpos - window)
        att = att.masked_fill(mask.unsqueeze(0), float("-inf"))
        if sinks is not None:
            att = torch.cat([sinks.float()[:, None, None].expand(QH, L, 1),
                             att], dim=-1)
            p = torch.softmax(att, dim=-1)[..., 1:]
        else:
            p = torch.softmax(att, dim=-1)
        o = torch.einsum("hqk,hkd->hqd", p, vs)
        out[s0:s1] = o.transpose(0, 1).to(q.dtype)
    return out


def context_attention(q: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, cu_seqlens_q: torch.Tensor,
                      kv_lens: torch.Tensor, block_tables: torch.Tensor,
                      scale: float, window: int = 0,
                      sinks=None) -> torch.Tensor:
    """Suffix-query causal attention over the paged cache. q: [Tq, QH, D];
    cache holds each sequence's FULL kv_len tokens (suffix included)."""
    Tq, QH, D = q.shape
    KH = k_cache.size(1)
    BS = k_cache.size(2)
    G = QH // KH
    out = torch.empty_like(q)
    cs = cu_seqlens_q.tolist()
    for b in range(len(cs) - 1):
        s0, s1 = cs[b], cs[b + 1]
        q_len = s1 - s0
        L = int(kv_lens[b])
        nb = (L + BS - 1) // BS
        blocks = block_tables[b, :nb].long()
        keys = k_cache[blocks].permute(1, 0, 2, 3).reshape(KH, nb * BS, D)[:, :L]
        vals = v_cache[blocks].permute(1, 0, 2, 3).reshape(KH, nb * BS, D)[:, :L]
        qs = q[s0:s1].float().transpose(0, 1)                  # [QH, q_len, D]
        kx = keys.float().repeat_interleave(G, 0)
        vx = vals.float().repeat_interleave(G, 0)
        att = torch.einsum("hqd,hkd->hqk", qs, kx) * scale
        qpos = torch.arange(L - q_len, L, device=q.device).unsqueeze(1)
        kpos = torch.arange(L, device=q.device).unsqueeze(0)
        mask = kpos > qpos
        if window > 0:
            mask = mask | (kpos <= qpos - window)
        att = att.masked_fill(mask.unsqueeze(0), float("-inf"))
        if sinks is not None:
            att = torch.cat(
                [sinks.float()[:, None, None].expand(QH, q_len, 1), att],
                dim=-1)
            p = torch.softmax(att, dim=-1)[..., 1:]
        else:
            p = torch.softmax(att, dim=-1)
        o = torch.einsum("hqk,hkd->hqd", p, vx)
        out[s0:s1] = o.transpose(0, 1).to(q.dtype)
    return out


def w4a16_unpack(qweight: torch.Tensor, scales: torch.Tensor,
                 zeros: torch.Tensor, group: int) -> torch.Tensor:
    """Dequantize the native W4A16 layout (qweight u32 [N, K/8], 8
    consecutive K nibbles per word; w = s*q - z) to float [N, K]."""
    N, kw = qweight.shape
    K = kw * 8
    shifts = torch.arange(8, device=qweight.device, dtype=torch.long) * 4
    q = (qweight.unsqueeze(-1).long() >> shifts) & 0xF      # [N, K/8, 8]
    q = q.reshape(N, K).float()
    s = scales.repeat_interleave(group, dim=1).float()       # [N, K]
    z = zeros.repeat_interleave(group, dim=1).float()
    return s * q - z


def w4a16_gemv(x: torch.Tensor, qweight: torch.Tensor, scales: torch.Tensor,
               zeros: torch.Tensor, group: int) -> torch.Tensor:
    w = w4a16_unpack(qweight, scales, zeros, group)
    return (x.float() @ w.T).to(x.dtype)


def mla_decode(q: torch.Tensor, cache: torch.Tensor,
               block_tables: torch.Tensor, seq_lens: torch.Tensor,
               scale: float, r: int) -> torch.Tensor:
    """Absorbed MLA decode reference (fp32 math). q: [T, H, r+rope] latent
    queries; cache: [NB, BS, r+rope] (c_kv || k_rope per token). Scores are
    the full latent dot; output is the latent-space V accumulation
    (first r dims). Mirrors ops/csrc/mla_attention.hip."""
    T, H, DT = q.shape
    NB, BS, _ = cache.shape
    out = torch.empty(T, H, r, dtype=q.dtype, device=q.device)
    for i in range(T):
        L = int(seq_lens[i])
        nb = (L + BS - 1) // BS
        rows = cache[block_tables[i, :nb].long()].reshape(-1, DT)[:L].float()
        s = q[i].float() @ rows.t() * scale          # [H, L]
        p = torch.softmax(s, dim=-1)
        out[i] = (p @ rows[:, :r]).to(out.dtype)
    return out
