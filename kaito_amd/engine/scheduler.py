"""Continuous-batching scheduler.

MI355X-native equivalent of the vLLM scheduler the reference drives through
--max-num-seqs and whose waiting-queue depth feeds the 429 rate limiter
(reference: presets/workspace/inference/vllm/rate_limit.py,
inference_api.py:645-658). Exposes num_waiting for the same guard.

Policy (v0-style, prefill-prioritized):
  * a step is either one PREFILL batch (token budget max_num_batched_tokens)
    or one DECODE batch over all running sequences;
  * decode preempts by evicting the newest sequence back to WAITING
    (recompute) when the block pool runs dry.
"""
from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field
from typing import List, Optional

from .block_pool import BlockPool
from .config import EngineConfig
from .sequence import Sequence, SeqStatus


@dataclass
class ScheduledBatch:
    is_prefill: bool
    seqs: List[Sequence] = field(default_factory=list)

    @property
    def num_seqs(self) -> int:
        return len(self.seqs)


class Scheduler:
    def __init__(self, cfg: EngineConfig, pool: BlockPool):
        self.cfg = cfg
        self.pool = pool
        self.waiting: deque[Sequence] = deque()
        self.running: List[Sequence] = []

    # ---- queue state (serves the rate limiter / metrics) ----
    @property
    def num_waiting(self) -> int:
        return len(self.waiting)

    @property
    def num_running(self) -> int:
        return len(self.running)

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def add(self, seq: Sequence) -> None:
        seq.status = SeqStatus.WAITING
        self.waiting.append(seq)

    # ---- scheduling ----
    def schedule(self) -> Optional[ScheduledBatch]:
        batch = self._schedule_prefill()
        if batch is not None:
            return batch
        return self._schedule_decode()

    def _schedule_prefill(self) -> Optional[ScheduledBatch]:
        if not self.waiting:
            return None
        budget = self.cfg.max_num_batched_tokens
        room = self.cfg.max_num_seqs - len(self.running)
        picked: List[Sequence] = []
        while self.waiting and room > 0:
            seq = self.waiting[0]
            n_tok = seq.num_prompt_tokens
            if picked and n_tok > budget:
                break
            need = self.pool.blocks_needed(seq.num_tokens + 1)
            if not self.pool.can_allocate(need):
                break
            self.waiting.popleft()
            seq.block_table = self.pool.allocate(need)
            seq.status = SeqStatus.RUNNING
            picked.append(seq)
            budget -= n_tok
            room -= 1
        if not picked:
            return None
        self.running.extend(picked)
        return ScheduledBatch(is_prefill=True, seqs=picked)

    def _schedule_decode(self) -> Optional[ScheduledBatch]:
        if not self.running:
            return None
        # skip seqs whose output is fully scheduled (pipelined step still in
        # flight — the engine resolves & finishes them)
        batch = [s for s in self.running
                 if s.sched_output_tokens < s.sampling.max_tokens]
        # grow block tables; preempt newest if pool dry
        i = 0
        while i < len(batch):
            seq = batch[i]
            need = self.pool.blocks_needed(seq.sched_tokens + 1) - len(seq.block_table)
            if need > 0:
                if self.pool.can_allocate(need):
                    seq.block_table.extend(self.pool.allocate(need))
                    seq._bt_dirty = True
                else:
                    victim = batch.pop()  # newest scheduled
                    self.running.remove(victim)
                    self.pool.free(victim.block_table)
                    victim.block_table = []
                    victim.output_token_ids = []
                    victim.sched_len = 0
                    victim.epoch += 1
                    victim.status = SeqStatus.WAITING
                    self.waiting.appendleft(victim)
                    continue  # retry same i (batch[i] is seq or its successor)
            i += 1
        if not batch:
            return None
        return ScheduledBatch(is_prefill=False, seqs=batch)

    def finish(self, seq: Sequence) -> None:
        self.pool.free(seq.block_table)
        seq.block_table = []
        if seq in self.running:
            self.running.remove(seq)
