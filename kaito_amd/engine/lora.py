"""Multi-adapter LoRA serving (punica/SGMV-class, reference row
"LoRA multi-adapter serving" SURVEY.md §2.3; adapters arrive via the
Workspace adapter spec + /mnt/adapter pulls, preset_inferences.go:886-956).

Adapters are stacked per target module into [num_slots, R_max, K] (A) and
[num_slots, O, R_max] (B) device tensors; per-token slot indices drive the
HIP shrink/expand kernels. CPU fallback mirrors the math for tests.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

# targets: module-suffix → (in_features key) handled generically
TARGET_SUFFIXES = ("qkv_proj", "o_proj", "gate_up_proj", "down_proj")

_ACTIVE_IDS: Optional[torch.Tensor] = None
_ACTIVE_MANAGER: Optional["LoRAManager"] = None


def set_active(manager: Optional["LoRAManager"],
               ids: Optional[torch.Tensor]) -> None:
    global _ACTIVE_IDS, _ACTIVE_MANAGER
    _ACTIVE_MANAGER = manager
    _ACTIVE_IDS = ids


def maybe_apply(module: torch.nn.Module, x: torch.Tensor,
                y: torch.Tensor) -> torch.Tensor:
    """Called by parallel linear layers after the base GEMM."""
    mgr = _ACTIVE_MANAGER
    if mgr is None or _ACTIVE_IDS is None:
        return y
    key = getattr(module, "_lora_key", None)
    if key is None or key not in mgr.stacks:
        return y
    return mgr.apply(key, x, y, _ACTIVE_IDS)


@dataclass
class LoRAAdapter:
    name: str
    rank: int
    alpha: float
    # module key → (A [r, in], B [out, r]) float tensors
    weights: Dict[str, tuple] = field(default_factory=dict)

    @property
    def scaling(self) -> float:
        return self.alpha / self.rank


class LoRAManager:
    def __init__(self, model: torch.nn.Module, max_adapters: int = 8,
                 max_rank: int = 64, device: str = "cpu",
                 dtype: torch.dtype = torch.bfloat16):
        self.device = torch.device(device)
        self.dtype = dtype
        self.max_adapters = max_adapters
        self.max_rank = max_rank
        self.slots: Dict[str, int] = {}
        self.scalings: List[float] = [1.0] * max_adapters
        # module key → dict(A=[N, R, K], B=[N, O, R])
        self.stacks: Dict[str, Dict[str, torch.Tensor]] = {}
        self._tag_modules(model)

    def _tag_modules(self, model: torch.nn.Module) -> None:
        self.module_shapes: Dict[str, tuple] = {}
        for name, mod in model.named_modules():
            if any(name.endswith(sfx) for sfx in TARGET_SUFFIXES) and \
                    hasattr(mod, "weight"):
                mod._lora_key = name
                o, k = mod.weight.shape
                self.module_shapes[name] = (k, o)

    def _ensure_stack(self, key: str):
        if key in self.stacks:
            return self.stacks[key]
        k, o = self.module_shapes[key]
        st = {
            "A": torch.zeros(self.max_adapters, self.max_rank, k,
                             dtype=self.dtype, device=self.device),
            "B": torch.zeros(self.max_adapters, o, self.max_rank,
                             dtype=self.dtype, device=self.device),
        }
        self.stacks[key] = st
        return st

    # ---------------------------------------------------------- registration
    def register(self, adapter: LoRAAdapter) -> int:
        if adapter.name in self.slots:
            slot = self.slots[adapter.name]
        else:
            if len(self.slots) >= self.max_adapters:
                raise RuntimeError("LoRA slots exhausted")
            slot = len(self.slots)
            self.slots[adapter.name] = slot
        self.scalings[slot] = adapter.scaling
        for key, (A, B) in adapter.weights.items():
            if key not in self.module_shapes:
                continue
            k, o = self.module_shapes[key]
            r = A.shape[0]
            if A.shape[1] != k or B.shape[0] != o or r > self.max_rank:
                continue  # adapter trained for a different base shape
            st = self._ensure_stack(key)
            # fold the adapter's alpha/r scaling into A once
            st["A"][slot, :r].copy_((A * adapter.scaling).to(self.dtype))
            st["A"][slot, r:].zero_()
            st["B"][slot, :, :r].copy_(B.to(self.dtype))
            st["B"][slot, :, r:].zero_()
        return slot

    def register_random(self, name: str, rank: int = 16, alpha: float = 32.0,
                        seed: int = 0, scale: float = 0.01) -> int:
        """Random adapter (tests / synthetic serving; no network)."""
        gen = torch.Generator().manual_seed(seed)
        weights = {}
        for key, (k, o) in self.module_shapes.items():
            A = torch.randn(rank, k, generator=gen) * scale
            B = torch.randn(o, rank, generator=gen) * scale
            weights[key] = (A, B)
        return self.register(LoRAAdapter(name, rank, alpha, weights))

    def load_peft_adapter(self, name: str, path: str) -> int:
        """Load a HF/peft adapter directory (adapter_config.json +
        adapter_model.safetensors) mapping q/k/v/gate/up splits onto our
        fused projections."""
        import json
        import os
        from safetensors.torch import safe_open
        with open(os.path.join(path, "adapter_config.json")) as f:
            cfg = json.load(f)
        rank = cfg.get("r", 16)
        alpha = cfg.get("lora_alpha", rank * 2)
        tensors = {}
        with safe_open(os.path.join(path, "adapter_model.safetensors"),
                       framework="pt", device="cpu") as sf:
            for k in sf.keys():
                tensors[k] = sf.get_tensor(k)

        def find(sub, ab):
            for k, v in tensors.items():
                if sub in k and f"lora_{ab}" in k:
                    return v.float()
            return None

        weights = {}
        for key, (kin, out) in self.module_shapes.items():
            lid = key.split(".")[1] if key.startswith("layers.") else None
            if key.endswith("qkv_proj"):
                parts = []
                ok = True
                for p in ("q_proj", "k_proj", "v_proj"):
                    A = find(f"layers.{lid}.self_attn.{p}", "A")
                    B = find(f"layers.{lid}.self_attn.{p}", "B")
                    if A is None or B is None:
                        ok = False
                        break
                    parts.append((A, B))
                if not ok:
                    continue
                # fused qkv: block-diagonal B over the q/k/v output ranges,
                # shared... simplest correct mapping: separate ranks stacked
                A = torch.cat([p[0] for p in parts], dim=0)          # [3r, k]
                Bs = [p[1] for p in parts]
                O = sum(b.shape[0] for b in Bs)
                B = torch.zeros(O, A.shape[0])
                ro = co = 0
                for b in Bs:
                    B[ro:ro + b.shape[0], co:co + rank] = b
                    ro += b.shape[0]
                    co += rank
                weights[key] = (A, B)
            elif key.endswith("gate_up_proj"):
                g = (find(f"layers.{lid}.mlp.gate_proj", "A"),
                     find(f"layers.{lid}.mlp.gate_proj", "B"))
                u = (find(f"layers.{lid}.mlp.up_proj", "A"),
                     find(f"layers.{lid}.mlp.up_proj", "B"))
                if any(x is None for x in g + u):
                    continue
                A = torch.cat([g[0], u[0]], dim=0)
                O = g[1].shape[0] + u[1].shape[0]
                B = torch.zeros(O, A.shape[0])
                B[:g[1].shape[0], :rank] = g[1]
                B[g[1].shape[0]:, rank:2 * rank] = u[1]
                weights[key] = (A, B)
            else:
                suffix = "self_attn.o_proj" if key.endswith("o_proj") else \
                    "mlp.down_proj"
                A = find(f"layers.{lid}.{suffix}", "A")
                B = find(f"layers.{lid}.{suffix}", "B")
                if A is not None and B is not None:
                    weights[key] = (A, B)
        return self.register(LoRAAdapter(name, rank, alpha, weights))

    def slot(self, name: Optional[str]) -> int:
        if name is None:
            return -1
        return self.slots.get(name, -1)

    # ---------------------------------------------------------- application
    def apply(self, key: str, x: torch.Tensor, y: torch.Tensor,
              ids: torch.Tensor) -> torch.Tensor:
        st = self.stacks[key]
        A, B = st["A"], st["B"]
        R = A.size(1)
        T = x.size(0)
        if x.is_cuda:
            from .. import ops
            ops.load_extension()
            tmp = torch.empty(T, R, dtype=torch.float32, device=x.device)
            # per-adapter alpha/r scaling is folded into A at registration
            torch.ops.kaito.lora_shrink(tmp, x, A, ids, 1.0)
            torch.ops.kaito.lora_expand(y, tmp, B, ids)
            return y
        # CPU reference path
        for t in range(T):
            a = int(ids[t])
            if a < 0:
                continue
            tmp = (A[a].float() @ x[t].float())
            y[t] += (B[a].float() @ tmp).to(y.dtype)
        return y

