"""Weight loading: HF-layout safetensors → our module tree.

Handles the fused projections (q/k/v → qkv_proj, gate/up → gate_up_proj)
and TP sharding. Random-init path lives on the model (bench contract:
no network, synthetic weights).
"""
from __future__ import annotations

from pathlib import Path
from typing import Dict

import torch

from ..parallel.state import get_state


def _shard(t: torch.Tensor, dim: int) -> torch.Tensor:
    st = get_state()
    if st.tp_size == 1:
        return t
    n = t.size(dim) // st.tp_size
    return t.narrow(dim, st.tp_rank * n, n).contiguous()


def load_safetensors_weights(model, path: str,
                             skip_projections: bool = False) -> None:
    from safetensors.torch import safe_open

    cfg = model.cfg
    files = sorted(Path(path).glob("*.safetensors"))
    if not files:
        raise FileNotFoundError(f"no safetensors under {path}")
    tensors: Dict[str, torch.Tensor] = {}
    for f in files:
        with safe_open(str(f), framework="pt", device="cpu") as sf:
            for k in sf.keys():
                tensors[k] = sf.get_tensor(k)

    def get(name):
        for prefix in ("model.", ""):
            if prefix + name in tensors:
                return tensors[prefix + name]
        raise KeyError(name)

    params = dict(model.named_parameters())

    def setp(name, value):
        with torch.no_grad():
            params[name].copy_(value.to(params[name].dtype))

    def has(name):
        return any(p + name in tensors for p in ("model.", ""))

    # norm weights: gemma stores w with y = x*(1+w); we fold to (1+w) at
    # load so the RMSNorm kernel is unchanged (cfg.rms_norm_offset)
    def norm_w(name):
        t = get(name)
        return t.float() + 1.0 if cfg.rms_norm_offset else t

    # head-dim padding (phi-2: checkpoint heads are 80-dim, the engine
    # runs the D∈{64,128,256} kernels on zero-padded 128-dim heads —
    # exact numerics, cfg.attn_scale keeps the 80^-0.5 scaling)
    def pad_heads(t, n_heads):
        d_ckpt = t.shape[0] // n_heads
        if d_ckpt == cfg.head_dim:
            return t
        x = t.reshape(n_heads, d_ckpt, -1)
        pad = torch.zeros(n_heads, cfg.head_dim - d_ckpt, x.shape[-1],
                          dtype=t.dtype)
        return torch.cat([x, pad], dim=1).reshape(n_heads * cfg.head_dim, -1)

    def pad_heads_cols(t, n_heads):
        d_ckpt = t.shape[1] // n_heads
        if d_ckpt == cfg.head_dim:
            return t
        x = t.reshape(t.shape[0], n_heads, d_ckpt)
        pad = torch.zeros(t.shape[0], n_heads, cfg.head_dim - d_ckpt,
                          dtype=t.dtype)
        return torch.cat([x, pad], dim=2).reshape(t.shape[0], -1)

    setp("embed_tokens.weight", _shard(get("embed_tokens.weight"), 0))
    # final norm (phi-2 checkpoints call it final_layernorm, with bias)
    final = "final_layernorm" if has("final_layernorm.weight") else "norm"
    setp("norm", norm_w(final + ".weight"))
    if getattr(model, "norm_bias", None) is not None and             has(final + ".bias"):
        setp("norm_bias", get(final + ".bias"))
    if model.lm_head is not None:
        setp("lm_head.weight", _shard(get("lm_head.weight"), 0))

    def attn_name(base, i):
        # phi-2 calls o_proj `dense`
        pre = f"layers.{i}.self_attn."
        if base == "o_proj" and not has(pre + "o_proj.weight") and                 has(pre + "dense.weight"):
            return pre + "dense"
        return pre + base

    # deepseek stores rope dims INTERLEAVED (HF apply_rotary permutes
    # inside the model); our rotate-half kernel wants the half-split
    # layout, so permute those output ROWS once at load — exact
    def deinterleave_pe(t, n_heads=1):
        d = t.shape[0] // n_heads
        x = t.reshape(n_heads, d // 2, 2, -1)
        return torch.cat([x[:, :, 0], x[:, :, 1]],
                         dim=1).reshape(t.shape[0], -1)

    def load_mla_attn(pre):
        a = pre + "self_attn."
        nope, rope, r = (cfg.qk_nope_head_dim, cfg.qk_rope_head_dim,
                         cfg.kv_lora_rank)
        if cfg.q_lora_rank > 0:
            setp(a + "q_a_proj", get(a + "q_a_proj.weight"))
            setp(a + "q_a_layernorm", get(a + "q_a_layernorm.weight"))
            qb = get(a + "q_b_proj.weight")
        else:
            qb = get(a + "q_proj.weight")
        # per-head [nope | rope] rows; de-interleave the rope part
        qb = qb.reshape(cfg.num_heads, nope + rope, -1)
        qb = torch.cat([qb[:, :nope],
                        deinterleave_pe(
                            qb[:, nope:].reshape(-1, qb.shape[-1]),
                            cfg.num_heads).reshape(cfg.num_heads, rope, -1)],
                       dim=1).reshape(cfg.num_heads * (nope + rope), -1)
        tgt = "q_b_proj" if cfg.q_lora_rank > 0 else "q_proj"
        setp(a + tgt + ".weight", _shard(qb, 0))
        kva = get(a + "kv_a_proj_with_mqa.weight")
        kva = torch.cat([kva[:r], deinterleave_pe(kva[r:])], 0)
        setp(a + "kv_a_proj_with_mqa", kva)
        setp(a + "kv_a_layernorm", get(a + "kv_a_layernorm.weight"))
        kvb = get(a + "kv_b_proj.weight").reshape(
            cfg.num_heads, nope + cfg.v_head_dim, r)
        setp(a + "w_kc", _shard(kvb[:, :nope], 0))
        setp(a + "w_vc", _shard(kvb[:, nope:].transpose(1, 2), 0))
        setp(a + "o_proj.weight", _shard(get(a + "o_proj.weight"), 1))

    # deepseek MoE: mlp.gate (+ e_score_correction_bias), per-expert
    # gate/up/down, optional shared_experts
    def load_deepseek_moe(pre, i):
        setp(pre + "mlp.gate", get(pre + "mlp.gate.weight"))
        if has(pre + "mlp.gate.e_score_correction_bias"):
            setp(pre + "mlp.e_score_correction_bias",
                 get(pre + "mlp.gate.e_score_correction_bias"))
        moe = params[pre + "mlp.w_gate_up"]
        e_base = getattr(model.layers[i].mlp, "e_base", 0)
        with torch.no_grad():
            for le in range(moe.shape[0]):
                ex = f"{pre}mlp.experts.{e_base + le}."
                params[pre + "mlp.w_gate_up"][le].copy_(torch.cat(
                    [get(ex + "gate_proj.weight"),
                     get(ex + "up_proj.weight")], 0).to(moe.dtype))
                params[pre + "mlp.w_down"][le].copy_(
                    get(ex + "down_proj.weight").to(moe.dtype))
        if cfg.n_shared_experts > 0:
            sh = pre + "mlp.shared_experts."
            setp(pre + "mlp.w_shared_gate_up", _shard(torch.cat(
                [get(sh + "gate_proj.weight"),
                 get(sh + "up_proj.weight")], 0), 0))
            setp(pre + "mlp.w_shared_down",
                 _shard(get(sh + "down_proj.weight"), 1))

    for i in range(cfg.num_layers):
        pre = f"layers.{i}."
        if cfg.is_mla and not skip_projections:
            load_mla_attn(pre)
            if i < cfg.first_k_dense or cfg.num_experts == 0:
                g = _shard(get(pre + "mlp.gate_proj.weight"), 0)
                u = _shard(get(pre + "mlp.up_proj.weight"), 0)
                setp(pre + "mlp.gate_up_proj.weight", torch.cat([g, u], 0))
                setp(pre + "mlp.down_proj.weight",
                     _shard(get(pre + "mlp.down_proj.weight"), 1))
            else:
                load_deepseek_moe(pre, i)
            setp(pre + "input_layernorm",
                 norm_w(pre + "input_layernorm.weight"))
            setp(pre + "post_attention_layernorm",
                 norm_w(pre + "post_attention_layernorm.weight"))
            continue
        if not skip_projections:   # AWQ checkpoints carry qweight instead
            q = pad_heads(get(attn_name("q_proj", i) + ".weight"),
                          cfg.num_heads)
            k = pad_heads(get(attn_name("k_proj", i) + ".weight"),
                          cfg.num_kv_heads)
            v = pad_heads(get(attn_name("v_proj", i) + ".weight"),
                          cfg.num_kv_heads)
            setp(pre + "self_attn.qkv_proj.weight", torch.cat(
                [_shard(q, 0), _shard(k, 0), _shard(v, 0)], 0))
            if cfg.attention_bias and has(attn_name("q_proj", i) + ".bias"):
                qb = pad_heads(get(attn_name("q_proj", i) + ".bias")
                               .unsqueeze(-1), cfg.num_heads).squeeze(-1)
                kb = pad_heads(get(attn_name("k_proj", i) + ".bias")
                               .unsqueeze(-1), cfg.num_kv_heads).squeeze(-1)
                vb = pad_heads(get(attn_name("v_proj", i) + ".bias")
                               .unsqueeze(-1), cfg.num_kv_heads).squeeze(-1)
                setp(pre + "self_attn.qkv_proj.bias", torch.cat(
                    [_shard(qb, 0), _shard(kb, 0), _shard(vb, 0)], 0))
            o = pad_heads_cols(get(attn_name("o_proj", i) + ".weight"),
                               cfg.num_heads)
            setp(pre + "self_attn.o_proj.weight", _shard(o, 1))
            if cfg.gated_mlp:
                g = _shard(get(pre + "mlp.gate_proj.weight"), 0)
                u = _shard(get(pre + "mlp.up_proj.weight"), 0)
                setp(pre + "mlp.gate_up_proj.weight", torch.cat([g, u], 0))
                setp(pre + "mlp.down_proj.weight",
                     _shard(get(pre + "mlp.down_proj.weight"), 1))
            else:   # phi-2/falcon ungated fc1/fc2
                fc1 = "mlp.fc1" if has(pre + "mlp.fc1.weight") else                     "mlp.up_proj"
                fc2 = "mlp.fc2" if has(pre + "mlp.fc2.weight") else                     "mlp.down_proj"
                setp(pre + "mlp.up_proj.weight",
                     _shard(get(pre + fc1 + ".weight"), 0))
                setp(pre + "mlp.down_proj.weight",
                     _shard(get(pre + fc2 + ".weight"), 1))
                if cfg.attention_bias and has(pre + fc1 + ".bias"):
                    setp(pre + "mlp.up_proj.bias",
                         _shard(get(pre + fc1 + ".bias"), 0))
                    setp(pre + "mlp.down_proj.bias",
                         get(pre + fc2 + ".bias"))
        if cfg.qk_norm:
            setp(pre + "self_attn.q_norm",
                 norm_w(pre + "self_attn.q_norm.weight"))
            setp(pre + "self_attn.k_norm",
                 norm_w(pre + "self_attn.k_norm.weight"))
        setp(pre + "input_layernorm", norm_w(pre + "input_layernorm.weight"))
        if cfg.norm_type == "layernorm" and                 has(pre + "input_layernorm.bias"):
            setp(pre + "input_layernorm_bias",
                 get(pre + "input_layernorm.bias"))
        if not cfg.parallel_block:
            setp(pre + "post_attention_layernorm",
                 norm_w(pre + "post_attention_layernorm.weight"))
        if cfg.sandwich_norms:
            setp(pre + "pre_feedforward_layernorm",
                 norm_w(pre + "pre_feedforward_layernorm.weight"))
            setp(pre + "post_feedforward_layernorm",
                 norm_w(pre + "post_feedforward_layernorm.weight"))
        # Mixtral-style MoE experts: block_sparse_moe.experts.N.w1/w3/w2
        if cfg.num_experts > 0 and                 has(pre + "block_sparse_moe.gate.weight"):
            setp(pre + "mlp.gate", get(pre + "block_sparse_moe.gate.weight"))
            moe = params[pre + "mlp.w_gate_up"]
            e_base = getattr(model.layers[i].mlp, "e_base", 0)
            for le in range(moe.shape[0]):
                ex = f"{pre}block_sparse_moe.experts.{e_base + le}."
                w1 = get(ex + "w1.weight")   # gate
                w3 = get(ex + "w3.weight")   # up
                with torch.no_grad():
                    params[pre + "mlp.w_gate_up"][le].copy_(
                        torch.cat([w1, w3], 0).to(moe.dtype))
                    params[pre + "mlp.w_down"][le].copy_(
                        get(ex + "w2.weight").to(moe.dtype))
