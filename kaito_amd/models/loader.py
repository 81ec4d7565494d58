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

    setp("embed_tokens.weight", _shard(get("embed_tokens.weight"), 0))
    setp("norm", get("norm.weight"))
    if model.lm_head is not None:
        setp("lm_head.weight", _shard(get("lm_head.weight"), 0))
    for i in range(cfg.num_layers):
        pre = f"layers.{i}."
        if not skip_projections:   # AWQ checkpoints carry qweight instead
            q = _shard(get(pre + "self_attn.q_proj.weight"), 0)
            k = _shard(get(pre + "self_attn.k_proj.weight"), 0)
            v = _shard(get(pre + "self_attn.v_proj.weight"), 0)
            setp(pre + "self_attn.qkv_proj.weight", torch.cat([q, k, v], 0))
            setp(pre + "self_attn.o_proj.weight",
                 _shard(get(pre + "self_attn.o_proj.weight"), 1))
            g = _shard(get(pre + "mlp.gate_proj.weight"), 0)
            u = _shard(get(pre + "mlp.up_proj.weight"), 0)
            setp(pre + "mlp.gate_up_proj.weight", torch.cat([g, u], 0))
            setp(pre + "mlp.down_proj.weight",
                 _shard(get(pre + "mlp.down_proj.weight"), 1))
        setp(pre + "input_layernorm", get(pre + "input_layernorm.weight"))
        setp(pre + "post_attention_layernorm",
             get(pre + "post_attention_layernorm.weight"))
