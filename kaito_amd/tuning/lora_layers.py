"""Trainable LoRA injection + int8 base quantization for fine-tuning.

The reference delegates to peft (get_peft_model) and bitsandbytes 4/8-bit
(presets/workspace/tuning/text-generation/fine_tuning.py:59-112); neither
ships in this image, so both are implemented natively on torch-ROCm:
  * LoRALinear: frozen base W (optionally int8 per-channel quantized) +
    trainable A/B with alpha/r scaling and dropout
  * inject_lora(): wraps target modules of a HF llama-class model
  * merge/save in peft-compatible adapter format so kaito_amd's serving
    LoRAManager.load_peft_adapter can load the result directly.
"""
from __future__ import annotations

import json
import math
import os
from typing import Dict, List

import torch
import torch.nn as nn

DEFAULT_TARGETS = ("q_proj", "k_proj", "v_proj", "o_proj",
                   "gate_proj", "up_proj", "down_proj")


class Int8Linear(nn.Module):
    """W8A16 frozen linear: per-output-channel symmetric int8 weights,
    dequantized on the fly (QLoRA base)."""

    def __init__(self, linear: nn.Linear):
        super().__init__()
        w = linear.weight.data.float()
        scale = w.abs().amax(dim=1, keepdim=True).clamp(min=1e-8) / 127.0
        self.register_buffer("qweight", torch.round(w / scale).to(torch.int8))
        self.register_buffer("scale", scale.to(torch.float32))
        self.bias = linear.bias
        self.in_features = linear.in_features
        self.out_features = linear.out_features

    def forward(self, x):
        w = (self.qweight.float() * self.scale).to(x.dtype)
        return nn.functional.linear(x, w, self.bias)


class LoRALinear(nn.Module):
    def __init__(self, base: nn.Module, rank: int, alpha: float,
                 dropout: float = 0.0, dtype=torch.float32):
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad_(False)
        in_f = base.in_features
        out_f = base.out_features
        self.rank = rank
        self.scaling = alpha / rank
        self.lora_A = nn.Parameter(torch.zeros(rank, in_f, dtype=dtype))
        self.lora_B = nn.Parameter(torch.zeros(out_f, rank, dtype=dtype))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        self.dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()

    def forward(self, x):
        y = self.base(x)
        xd = self.dropout(x)
        delta = (xd.to(self.lora_A.dtype) @ self.lora_A.T) @ self.lora_B.T
        return y + self.scaling * delta.to(y.dtype)


def inject_lora(model: nn.Module, rank: int = 16, alpha: float = 32.0,
                dropout: float = 0.05, targets=DEFAULT_TARGETS,
                quantize_base: bool = False) -> List[str]:
    """Wraps matching nn.Linear modules in-place; freezes everything else.
    Returns the list of wrapped module names."""
    for p in model.parameters():
        p.requires_grad_(False)
    wrapped = []
    for name, mod in list(model.named_modules()):
        leaf = name.split(".")[-1]
        if leaf in targets and isinstance(mod, nn.Linear):
            parent = model.get_submodule(name.rsplit(".", 1)[0]) if "." in name \
                else model
            base = Int8Linear(mod) if quantize_base else mod
            setattr(parent, leaf, LoRALinear(base, rank, alpha, dropout))
            wrapped.append(name)
    return wrapped


def trainable_parameters(model: nn.Module):
    return [p for p in model.parameters() if p.requires_grad]


def save_adapter(model: nn.Module, out_dir: str, rank: int, alpha: float,
                 base_model_name: str = "", targets=DEFAULT_TARGETS) -> None:
    """peft-compatible adapter output (adapter_config.json +
    adapter_model.safetensors) + the completion marker the reference's
    pusher sidecar waits on (fine_tuning.py:155-166)."""
    from safetensors.torch import save_file
    os.makedirs(out_dir, exist_ok=True)
    tensors: Dict[str, torch.Tensor] = {}
    for name, mod in model.named_modules():
        if isinstance(mod, LoRALinear):
            tensors[f"base_model.model.{name}.lora_A.weight"] = \
                mod.lora_A.data.float().contiguous()
            tensors[f"base_model.model.{name}.lora_B.weight"] = \
                mod.lora_B.data.float().contiguous()
    save_file(tensors, os.path.join(out_dir, "adapter_model.safetensors"))
    with open(os.path.join(out_dir, "adapter_config.json"), "w") as f:
        json.dump({
            "peft_type": "LORA",
            "r": rank,
            "lora_alpha": alpha,
            "target_modules": list(targets),
            "base_model_name_or_path": base_model_name,
            "task_type": "CAUSAL_LM",
        }, f, indent=2)
    # completion marker (reference writes 'fine_tuning_completed.txt')
    with open(os.path.join(out_dir, "fine_tuning_completed.txt"), "w") as f:
        f.write("done\n")
