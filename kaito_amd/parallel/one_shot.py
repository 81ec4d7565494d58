"""One-shot fused all-reduce + RMSNorm over xGMI peer mappings.

SURVEY.md hard part #2: on the 8-GPU MI355X mesh every GPU has a direct
xGMI link to every peer, so for decode-sized tensors a one-shot
all-reduce (each rank reads all N peer buffers and reduces locally)
replaces the per-link-bound ring, and fusing the RMSNorm epilogue saves
one full read+write of the hidden states (the fusion the reference
explicitly disables on NVIDIA, interface.go:439-446).

OneShotGroup wires the cross-process plumbing: each rank registers a
staging buffer, exchanges hipIpc handles over torch.distributed
(all_gather_object), opens its peers' mappings, and then every
`allreduce_rmsnorm(x, weight)` is: copy x into the staging buffer,
barrier, one fused kernel. Requires one process per GPU on one node with
dmabuf IPC (HSA_ENABLE_IPC_MODE_LEGACY=0 — the pool's default).

Single-GPU tests exercise the kernel itself through `fused_local` with N
local buffers; the IPC path needs a multi-GPU node (round-2 validation,
docs/ROADMAP.md).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from .. import ops


def fused_local(inputs: List[torch.Tensor], weight: torch.Tensor,
                eps: float) -> torch.Tensor:
    """Fused sum(inputs) → RMSNorm, all buffers local (the kernel is
    identical for IPC-mapped peers; this is the testable core)."""
    if not inputs[0].is_cuda:
        acc = torch.zeros_like(inputs[0], dtype=torch.float32)
        for t in inputs:
            acc += t.float()
        var = acc.pow(2).mean(-1, keepdim=True)
        return (acc * torch.rsqrt(var + eps) *
                weight.float()).to(inputs[0].dtype)
    ops.load_extension()
    out = torch.empty_like(inputs[0])
    ptrs = torch.tensor([t.data_ptr() for t in inputs], dtype=torch.long,
                        device=inputs[0].device)
    torch.ops.kaito.allreduce_rmsnorm(out, ptrs, weight, eps)
    return out


class OneShotGroup:
    """Cross-process one-shot group (one rank per GPU, same node)."""

    def __init__(self, max_tokens: int, hidden: int,
                 dtype: torch.dtype = torch.bfloat16):
        import torch.distributed as dist
        self.dist = dist
        self.world = dist.get_world_size()
        self.rank = dist.get_rank()
        self.buf = torch.empty(max_tokens, hidden, dtype=dtype,
                               device="cuda")
        ops.load_extension()
        handle = torch.ops.kaito.ipc_handle(self.buf).cpu()
        handles: List[Optional[torch.Tensor]] = [None] * self.world
        dist.all_gather_object(handles, handle)
        ptrs = []
        self._opened = []
        for r, h in enumerate(handles):
            if r == self.rank:
                ptrs.append(self.buf.data_ptr())
            else:
                p = torch.ops.kaito.ipc_open(h)
                ptrs.append(p)
                self._opened.append(p)
        self.ptrs = torch.tensor(ptrs, dtype=torch.long, device="cuda")

    def allreduce_rmsnorm(self, x: torch.Tensor, weight: torch.Tensor,
                          eps: float) -> torch.Tensor:
        """x: [T, H] local partial; returns rmsnorm(sum over ranks)."""
        T = x.size(0)
        self.buf[:T].copy_(x)
        torch.cuda.synchronize()
        self.dist.barrier()          # peers' staging writes visible
        out = torch.empty_like(x)
        torch.ops.kaito.allreduce_rmsnorm(out, self.ptrs, weight, eps)
        torch.cuda.synchronize()
        self.dist.barrier()          # nobody overwrites staging early
        return out

    def close(self):
        for p in self._opened:
            torch.ops.kaito.ipc_close(p)
        self._opened = []
