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
class PrefillChunk:
    seq: Sequence
    start: int          # first prompt token index in this chunk
    length: int

    @property
    def completes(self) -> bool:
        # context (prompt + preserved output after a recompute preemption)
        return self.start + self.length >= self.seq.num_context_tokens


@dataclass
class ScheduledBatch:
    is_prefill: bool
    seqs: List[Sequence] = field(default_factory=list)
    # prefill batches: per-seq chunk ranges (chunked prefill); seqs holds
    # the chunk owners in order
    chunks: List[PrefillChunk] = field(default_factory=list)

    @property
    def num_seqs(self) -> int:
        return len(self.seqs)

    @property
    def sampling_seqs(self) -> List[Sequence]:
        """Sequences that get a token sampled this step (decode: all;
        prefill: only chunks completing their prompt)."""
        if not self.is_prefill:
            return self.seqs
        return [c.seq for c in self.chunks if c.completes]


class Scheduler:
    def __init__(self, cfg: EngineConfig, pool: BlockPool,
                 restore_cb=None):
        self.cfg = cfg
        self.pool = pool
        self.waiting: deque[Sequence] = deque()
        self.prefilling: List[Sequence] = []   # admitted, prompt KV partial
        self.running: List[Sequence] = []
        # called at admission: restore_cb(seq) -> covered prefix tokens
        self.restore_cb = restore_cb
        # opt-in: admit only prompts that fit the step budget whole
        # (kept as a scheduling mode; MLA no longer needs it — the
        # latent-cache context path handles suffix chunks)
        self.whole_prompt_only = False

    # ---- queue state (serves the rate limiter / metrics) ----
    @property
    def num_waiting(self) -> int:
        return len(self.waiting)

    @property
    def num_running(self) -> int:
        return len(self.running) + len(self.prefilling)

    def has_work(self) -> bool:
        return bool(self.waiting or self.prefilling or self.running)

    def add(self, seq: Sequence) -> None:
        seq.status = SeqStatus.WAITING
        self.waiting.append(seq)

    # ---- scheduling ----
    def schedule(self) -> Optional[ScheduledBatch]:
        batch = self._schedule_prefill()
        if batch is not None:
            return batch
        return self._schedule_decode()

    def schedule_mixed(self, prefill_budget: int,
                       allow_prefill: bool = True):
        """Decode-priority mixed scheduling: a decode batch over all running
        sequences PLUS a bounded prefill chunk the engine overlaps on a side
        stream (allow_prefill gates it while a prior chunk is in flight).
        Falls back to full-budget pure prefill when nothing is decoding
        (startup burst)."""
        decode = self._schedule_decode()
        if decode is None:
            return None, self._schedule_prefill()
        if not allow_prefill:
            return decode, None
        return decode, self._schedule_prefill(budget=prefill_budget)

    def _schedule_prefill(self, budget: Optional[int] = None
                          ) -> Optional[ScheduledBatch]:
        """Chunked prefill: each step processes up to max_num_batched_tokens
        of prompt tokens; long prompts span multiple steps (context
        attention handles suffix chunks). New sequences are admitted after
        in-flight prefills continue."""
        if not self.waiting and not self.prefilling:
            return None
        if budget is None:
            budget = self.cfg.max_num_batched_tokens
        chunks: List[PrefillChunk] = []

        # 1. continue partially-prefilled sequences (remaining == 0 means
        # the seq's last chunk is in flight, awaiting promotion)
        for seq in list(self.prefilling):
            if budget <= 0:
                break
            remaining = seq.num_context_tokens - seq.prefilled_len
            if remaining <= 0:
                continue
            n = min(remaining, budget)
            chunks.append(PrefillChunk(seq, seq.prefilled_len, n))
            budget -= n

        # 2. admit new sequences (blocks for the whole prompt upfront)
        room = self.cfg.max_num_seqs - len(self.running) - len(self.prefilling)
        while self.waiting and room > 0 and budget > 0:
            seq = self.waiting[0]
            need = self.pool.blocks_needed(seq.num_tokens + 1)
            if not self.pool.can_allocate(need):
                break
            self.waiting.popleft()
            shared = []
            if hasattr(self.pool, "match_prefix"):
                # automatic prefix caching: skip prefill for cached full
                # prompt blocks (KV already resident, zero copies)
                shared, covered = self.pool.match_prefix(
                    seq.prompt_token_ids)
                seq.prefilled_len = max(seq.prefilled_len, covered)
            seq.block_table = shared + self.pool.allocate(need - len(shared))
            seq.status = SeqStatus.RUNNING
            if self.restore_cb is not None and seq.prefilled_len == 0:
                seq.prefilled_len = self.restore_cb(seq)
            if seq.prefilled_len >= seq.num_context_tokens:
                # full prefix-cache hit: straight to decode (the next decode
                # step feeds the last context token over restored KV)
                seq.sched_len = seq.num_context_tokens
                self.running.append(seq)
                room -= 1
                continue
            remaining = seq.num_context_tokens - seq.prefilled_len
            if self.whole_prompt_only and remaining > budget:
                # put it back; it needs a step with a larger free budget
                self.waiting.appendleft(seq)
                self.pool.free(seq.block_table)
                seq.block_table = []
                seq.status = SeqStatus.WAITING
                break
            n = min(remaining, budget)
            chunks.append(PrefillChunk(seq, seq.prefilled_len, n))
            self.prefilling.append(seq)
            budget -= n
            room -= 1

        if not chunks:
            return None
        # completion bookkeeping happens in the engine after execution
        return ScheduledBatch(is_prefill=True,
                              seqs=[c.seq for c in chunks], chunks=chunks)

    def finish_prefill_chunks(self, batch: ScheduledBatch) -> None:
        """Advance prefilled_len; move completed sequences to running."""
        self.advance_prefill_chunks(batch)
        self.promote_prefilled([c.seq for c in batch.chunks if c.completes])

    def advance_prefill_chunks(self, batch: ScheduledBatch) -> None:
        """Mark chunk tokens as (about to be) in the cache — called at
        LAUNCH so the next step schedules the following chunk, not a
        repeat."""
        for c in batch.chunks:
            c.seq.prefilled_len = c.start + c.length

    def promote_prefilled(self, seqs: List[Sequence]) -> None:
        """Move fully-prefilled sequences into the decode set — called at
        RESOLVE (fire-and-forget prefill: a seq may only decode after its
        prefill KV writes are known complete on the GPU)."""
        for s in seqs:
            if s in self.prefilling:
                self.prefilling.remove(s)
                self.running.append(s)

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
                    # recompute preemption (vLLM semantics): KEEP the
                    # generated tokens — they were already streamed to the
                    # client — and re-prefill prompt+generated as context
                    # at re-admission (prefill paths use context_token_ids)
                    victim.block_table = []
                    victim.sched_len = 0
                    victim.prefilled_len = 0
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
        if seq in self.prefilling:
            self.prefilling.remove(seq)
