"""Distributed process-group state: one process per GPU, RCCL over xGMI.

On ROCm the torch.distributed "nccl" backend IS RCCL; xGMI gives every
MI355X a direct link to every other GPU in the 8-way mesh, so TP
all-reduces run as single-hop collectives. CPU tests use gloo.

Reference behaviour being replaced: vLLM's NCCL TP groups + Ray bootstrap
(SURVEY.md §5.8); rendezvous here is plain torch.distributed env-var init
(POD_INDEX/headless-service DNS on k8s, 127.0.0.1 standalone).
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class ParallelState:
    world_size: int = 1
    rank: int = 0
    local_rank: int = 0
    tp_size: int = 1
    tp_rank: int = 0
    tp_group: Optional[object] = None    # ProcessGroup
    dp_size: int = 1
    dp_rank: int = 0
    # pipeline parallel (tier 3): ranks laid out TP-minor —
    # rank = pp_rank * tp_size + tp_rank
    pp_size: int = 1
    pp_rank: int = 0

    @property
    def is_first_stage(self) -> bool:
        return self.pp_rank == 0

    @property
    def is_last_stage(self) -> bool:
        return self.pp_rank == self.pp_size - 1

    @property
    def prev_stage_rank(self) -> int:
        return self.rank - self.tp_size

    @property
    def next_stage_rank(self) -> int:
        return self.rank + self.tp_size


_STATE = ParallelState()


def get_state() -> ParallelState:
    return _STATE


def is_initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


def init_parallel(tp_size: int = 1, pp_size: int = 1,
                  backend: Optional[str] = None,
                  timeout_s: int = 600) -> ParallelState:
    """Initialise torch.distributed (if WORLD_SIZE>1) and carve TP/PP
    groups. Ranks are TP-minor: rank = (dp*pp_size + pp)*tp_size + tp; TP
    groups are contiguous rank ranges; pipeline stages are tp_size apart."""
    global _STATE
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        import datetime
        dist.init_process_group(backend=backend,
                                timeout=datetime.timedelta(seconds=timeout_s))

    tp_size = max(1, min(tp_size, world))
    pp_size = max(1, min(pp_size, world // tp_size))
    assert world % (tp_size * pp_size) == 0,         f"world {world} not divisible by tp*pp {tp_size * pp_size}"
    tp_group = None
    tp_rank = 0
    if world > 1:
        n_groups = world // tp_size
        for g in range(n_groups):
            ranks = list(range(g * tp_size, (g + 1) * tp_size))
            grp = dist.new_group(ranks) if tp_size > 1 else None
            if rank in ranks:
                tp_group = grp
                tp_rank = rank - g * tp_size
    pp_rank = (rank // tp_size) % pp_size
    _STATE = ParallelState(
        world_size=world, rank=rank, local_rank=local_rank,
        tp_size=tp_size, tp_rank=tp_rank, tp_group=tp_group,
        dp_size=world // (tp_size * pp_size),
        dp_rank=rank // (tp_size * pp_size),
        pp_size=pp_size, pp_rank=pp_rank)
    return _STATE


def pp_send_next(t: torch.Tensor) -> None:
    dist.send(t.contiguous(), _STATE.next_stage_rank)


def pp_recv_prev(shape, dtype, device) -> torch.Tensor:
    t = torch.empty(*shape, dtype=dtype, device=device)
    dist.recv(t, _STATE.prev_stage_rank)
    return t


def pp_broadcast_from_last(t: torch.Tensor) -> torch.Tensor:
    """Broadcast a tensor from the LAST pipeline stage (tp rank 0 of that
    stage) to all ranks — the sampled token ids each step."""
    src = (_STATE.pp_size - 1) * _STATE.tp_size
    dist.broadcast(t, src)
    return t


def tp_all_reduce(t: torch.Tensor) -> torch.Tensor:
    st = _STATE
    if st.tp_size > 1:
        dist.all_reduce(t, group=st.tp_group)
    return t


def tp_all_gather(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    st = _STATE
    if st.tp_size == 1:
        return t
    parts = [torch.empty_like(t) for _ in range(st.tp_size)]
    dist.all_gather(parts, t, group=st.tp_group)
    return torch.cat(parts, dim=dim)


def barrier() -> None:
    if is_initialized():
        dist.barrier()


def destroy() -> None:
    global _STATE
    if dist.is_initialized():
        dist.destroy_process_group()
    _STATE = ParallelState()
