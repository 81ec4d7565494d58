"""One-shot fused all-reduce(+residual)+RMSNorm over xGMI peer mappings.

SURVEY.md hard part #2: on the 8-GPU MI355X mesh every GPU has a direct
xGMI link to every peer, so for decode-sized tensors a one-shot
all-reduce (each rank reads all N peer staging buffers and reduces
locally) replaces the per-link-bound ring, and fusing the
residual-add + RMSNorm epilogue saves one full read+write of the hidden
states (the fusion the reference explicitly disables on NVIDIA,
interface.go:439-446).

v2 (round 2): the synchronization lives INSIDE the kernel — each rank
exchanges hipIpc handles for a staging pair AND a signal page; block
`row` release-stores a device-side epoch into every peer's signal slot
and acquire-spins until all peers arrive. With double-buffered staging
(a peer that entered call k+1 has necessarily finished reading call k's
buffer: kernels on one stream serialize) one in-kernel barrier per call
suffices — no host-side dist.barrier, no torch.cuda.synchronize — so
the whole TP decode step, collectives included, captures into a
hipGraph.

Model wiring: `activate(group)` installs the group; RowParallelLinear
skips its NCCL all-reduce when `defer(T)` says the following
fused_add_rms_norm call site will do the reduction
(models/llama.py decoder layer + final norm). Prefill batches larger
than the staging window fall back to the plain RCCL ring all-reduce —
one-shot's N-times read amplification only wins at decode sizes.

The hipIpc path needs one process per GPU on one node
(HSA_ENABLE_IPC_MODE_LEGACY=0 — the pool's default). CPU/gloo test
processes use GlooEmulatedGroup, which reproduces the exact semantics
(all-reduce → bf16 residual add → RMSNorm) with torch collectives so
the engine-level wiring is validated by world-2 tests here.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from .. import ops
from .state import get_state


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
    """Cross-process one-shot group (one rank per GPU, same node).

    allreduce_add_rmsnorm() is graph-capturable: staging copy + one
    kernel, no host synchronization.
    """

    def __init__(self, max_tokens: int, hidden: int,
                 dtype: torch.dtype = torch.bfloat16):
        import torch.distributed as dist
        self.dist = dist
        st = get_state()
        self.world = st.tp_size
        self.rank = st.tp_rank
        self.group = st.tp_group
        self.max_tokens = max_tokens
        self.hidden = hidden
        ops.load_extension()
        # double-buffered staging in ONE allocation (one IPC handle)
        self.buf = torch.empty(2, max_tokens, hidden, dtype=dtype,
                               device="cuda")
        self._buf_stride = max_tokens * hidden * self.buf.element_size()
        # signal page: [max_tokens rows][8 peer slots] u32 + local epochs
        self.sig = torch.zeros(max_tokens * 8, dtype=torch.int32,
                               device="cuda")
        self.counter = torch.zeros(max_tokens, dtype=torch.int32,
                                   device="cuda")
        buf_ptrs, self._opened_buf = self._exchange(self.buf)
        sig_ptrs, self._opened_sig = self._exchange(self.sig)
        dev = "cuda"
        self.ptrs = [
            torch.tensor(buf_ptrs, dtype=torch.long, device=dev),
            torch.tensor([p + self._buf_stride for p in buf_ptrs],
                         dtype=torch.long, device=dev),
        ]
        self.sig_ptrs = torch.tensor(sig_ptrs, dtype=torch.long, device=dev)
        self._flip = 0

    def _exchange(self, t: torch.Tensor):
        handle = torch.ops.kaito.ipc_handle(t).cpu()
        handles: List[Optional[torch.Tensor]] = [None] * self.world
        self.dist.all_gather_object(handles, handle, group=self.group)
        ptrs, opened = [], []
        for r, h in enumerate(handles):
            if r == self.rank:
                ptrs.append(t.data_ptr())
            else:
                p = torch.ops.kaito.ipc_open(h)
                ptrs.append(p)
                opened.append(p)
        return ptrs, opened

    def begin_step(self) -> None:
        """Reset the staging flip at a deterministic point (start of each
        model forward) so hipGraph capture and replay see the same
        buffer sequence."""
        self._flip = 0

    def allreduce_add_rmsnorm(self, x: torch.Tensor, residual: torch.Tensor,
                              weight: torch.Tensor, eps: float):
        """x: [T, H] rank-local partial. residual updated IN PLACE to
        residual + sum(x over ranks) (bf16 stream); returns
        (rmsnorm(residual), residual). Pass residual=empty for the plain
        allreduce+norm (final PP stage boundary)."""
        T = x.size(0)
        buf = self.buf[self._flip]
        ptrs = self.ptrs[self._flip]
        self._flip ^= 1
        buf[:T].copy_(x.view(T, -1))
        out = torch.empty_like(x)
        torch.ops.kaito.one_shot_ar_rmsnorm(
            out.view(T, -1), residual.view(T, -1) if residual.numel() else
            residual, ptrs, self.sig_ptrs, self.counter, weight, eps,
            self.rank)
        return out, residual

    # v1 API (host-barrier path) kept for the existing numerics tests
    def allreduce_rmsnorm(self, x: torch.Tensor, weight: torch.Tensor,
                          eps: float) -> torch.Tensor:
        empty = torch.empty(0, dtype=x.dtype, device=x.device)
        out, _ = self.allreduce_add_rmsnorm(x, empty, weight, eps)
        return out

    def close(self):
        for p in self._opened_buf + self._opened_sig:
            torch.ops.kaito.ipc_close(p)
        self._opened_buf = []
        self._opened_sig = []


class GlooEmulatedGroup:
    """Semantics-equivalent group for CPU/gloo world-N tests: all-reduce
    with torch.distributed, then the torch_ref bf16 residual-add +
    RMSNorm. Validates the engine/model wiring (deferred RowParallel
    reduce + fused call sites) without GPUs."""

    def __init__(self, max_tokens: int, hidden: int,
                 dtype: torch.dtype = torch.bfloat16):
        import torch.distributed as dist
        self.dist = dist
        st = get_state()
        self.group = st.tp_group
        self.max_tokens = max_tokens

    def begin_step(self) -> None:
        pass

    def allreduce_add_rmsnorm(self, x: torch.Tensor, residual: torch.Tensor,
                              weight: torch.Tensor, eps: float):
        from ..ops import torch_ref
        red = x.float()
        self.dist.all_reduce(red, group=self.group)
        red = red.to(x.dtype)
        if residual.numel():
            out, res = torch_ref.fused_add_rms_norm(red, residual, weight,
                                                    eps)
            residual.copy_(res)
            return out, residual
        return torch_ref.rms_norm(red, weight, eps), residual

    def close(self):
        pass


_ACTIVE = None


def activate(group) -> None:
    global _ACTIVE
    _ACTIVE = group


def active():
    return _ACTIVE


def deactivate() -> None:
    global _ACTIVE
    if _ACTIVE is not None:
        _ACTIVE.close()
    _ACTIVE = None


def defer(num_tokens: int):
    """Returns the active group when the upcoming row-parallel output of
    `num_tokens` rows should SKIP its ring all-reduce (the fused one-shot
    call site will reduce instead); None → reduce normally."""
    g = _ACTIVE
    if g is not None and num_tokens <= g.max_tokens:
        return g
    return None


def make_group(max_tokens: int, hidden: int,
               dtype: torch.dtype = torch.bfloat16):
    """OneShotGroup on GPU (hipIpc + RCCL), GlooEmulatedGroup otherwise."""
    if torch.cuda.is_available():
        return OneShotGroup(max_tokens, hidden, dtype)
    return GlooEmulatedGroup(max_tokens, hidden, dtype)
