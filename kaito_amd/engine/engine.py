"""LLMEngine: the continuous-batching serving loop.

The MI355X-native replacement for the vLLM engine the reference's Workspace
pods launch (SURVEY.md §3.2 "HOT LOOP"). One engine per GPU (DP tier) or per
TP group; step() = schedule → forward → sample → append/finish.
"""
from __future__ import annotations

import itertools
import logging
import time
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch

from .config import EngineConfig
from .block_pool import BlockPool
from .model_runner import ModelRunner
from .sampler import Sampler
from .scheduler import Scheduler
from .sequence import SamplingParams, Sequence, SeqStatus

logger = logging.getLogger(__name__)


@dataclass
class _PendingStep:
    seqs: List[Sequence]
    epochs: List[int]
    tokens: torch.Tensor                 # [bs] device (feeds the next step)
    host: Optional[torch.Tensor]         # pinned copy (GPU) or None (CPU)
    event: Optional[object]              # cuda Event or None
    index: Dict[int, int]                # seq_id -> batch index


class LLMEngine:
    def __init__(self, cfg: EngineConfig, weights_path: Optional[str] = None):
        self.cfg = cfg
        self.runner = ModelRunner(cfg).load_model(weights_path, cfg.seed)
        self.runner.setup_tunable()
        self.runner.profile_and_allocate_kv()
        self.pool = BlockPool(self.runner.num_gpu_blocks, cfg.block_size)
        self.kv_offload = None
        if cfg.kv_offload:
            from .kv_offload import KVOffloadManager
            self.kv_offload = KVOffloadManager(
                self.runner.kv_caches, cfg.block_size,
                cfg.kv_offload_bytes, cfg.device)
        restore_cb = None
        if self.kv_offload is not None:
            restore_cb = lambda seq: self.kv_offload.restore_prefix(  # noqa: E731
                seq.prompt_token_ids, seq.block_table)
        self.scheduler = Scheduler(cfg, self.pool, restore_cb)
        self.sampler = Sampler(cfg.device, cfg.seed)
        from ..parallel.state import get_state
        self._pp = get_state().pp_size > 1
        self.eos_token_id: Optional[int] = None
        self._next_id = itertools.count()
        self.seqs: Dict[int, Sequence] = {}
        self._pending: Optional[_PendingStep] = None
        # counters for /metrics (names consumed by the benchmark probe)
        self.num_generation_tokens = 0
        self.num_prompt_tokens = 0
        # optional KV event bus (EPP KVCache-aware routing surface)
        self.kv_publisher = None

    def capture_graphs(self):
        self.runner.capture_decode_graphs()
        return self

    # ------------------------------------------------------------- requests
    def add_request(self, prompt_token_ids: List[int],
                    sampling: Optional[SamplingParams] = None,
                    seq_id: Optional[int] = None,
                    lora_name: Optional[str] = None) -> int:
        sid = seq_id if seq_id is not None else next(self._next_id)
        seq = Sequence(sid, list(prompt_token_ids),
                       sampling or SamplingParams())
        if lora_name is not None:
            mgr = self.runner.lora_manager
            if mgr is None:
                raise ValueError("LoRA not enabled (EngineConfig.enable_lora)")
            seq.lora_id = mgr.slot(lora_name)
            if seq.lora_id < 0:
                raise KeyError(f"unknown LoRA adapter {lora_name!r}")
        self.seqs[sid] = seq
        self.scheduler.add(seq)
        return sid

    def abort(self, seq_id: int) -> None:
        seq = self.seqs.pop(seq_id, None)
        if seq is None:
            return
        if seq.status == SeqStatus.RUNNING:
            self.scheduler.finish(seq)
        elif seq in self.scheduler.waiting:
            self.scheduler.waiting.remove(seq)
        seq.status = SeqStatus.FINISHED
        seq.finish_reason = "abort"

    def has_unfinished(self) -> bool:
        return self.scheduler.has_work() or self._pending is not None

    # ------------------------------------------------------------- stepping
    #
    # Pipelined decode: step N is LAUNCHED before step N-1's sampled tokens
    # reach the host. The GPU stays busy through the host-side bookkeeping
    # that used to sit between steps (~ms per step at bs=256). Token VALUES
    # flow device-side between steps (runner fast path / pending_map);
    # host bookkeeping (append, EOS/length finish) happens one step late at
    # _resolve_pending(). A sequence that emits EOS costs one speculative
    # extra decode step (its KV write lands in its own still-allocated spare
    # block — scheduler pre-allocates +1 token, so it's harmless).
    @torch.no_grad()
    def step(self) -> List[Sequence]:
        """One engine iteration. Returns sequences that FINISHED this step."""
        if self.cfg.enable_mixed_batch and not self._pp:
            return self._step_mixed()
        return self._step_classic()

    def _prefill_bookkeep(self, batch) -> List[Sequence]:
        """Post-execution prefill bookkeeping (shared by the sync and mixed
        paths): advance chunks, set sched_len, count prompt tokens, publish
        KV events. Returns the sampling seqs."""
        self.scheduler.finish_prefill_chunks(batch)
        samp = batch.sampling_seqs
        for seq in samp:
            seq.sched_len = seq.num_prompt_tokens + 1
            self.num_prompt_tokens += seq.num_prompt_tokens
        if self.kv_publisher is not None and samp:
            blocks = [b for s2 in samp for b in s2.block_table]
            self.kv_publisher.block_stored(blocks)
        return samp

    def _side_stream(self):
        s = getattr(self, "_side", None)
        if s is None:
            s = self._side = torch.cuda.Stream()
        return s

    @torch.no_grad()
    def _step_mixed(self) -> List[Sequence]:
        """Mixed step: decode over all running seqs on the main stream
        (hipGraph), with a bounded prefill chunk launched CONCURRENTLY on a
        side stream. Prefill's MFMA-bound GEMMs overlap decode's
        bandwidth-bound paged attention — on the bench workload pure
        prefill steps are ~25-30% of wall time, and this hides them.

        Ordering: the side stream waits on an event recorded BEFORE this
        step's decode launch. That event orders the prefill after (a) the
        previous step's decode KV writes — whose target blocks may have
        been freed by a finish/preemption and re-allocated to a prefill
        seq this step — and (b) this step's prefix-restore H2D copies,
        without serializing it after this step's decode."""
        d_batch, p_batch = self.scheduler.schedule_mixed(
            self.cfg.mixed_prefill_tokens)
        if d_batch is None and p_batch is None:
            return self._resolve_pending()
        finished: List[Sequence] = []
        if d_batch is None:
            # startup burst: classic synchronous full-budget prefill
            finished += self._resolve_pending()
            hidden = self.runner.execute_prefill(p_batch.chunks)
            samp = self._prefill_bookkeep(p_batch)
            if not samp:
                return finished
            tokens = self._sample_maybe_pp(hidden, samp)
            finished += self._commit(samp, tokens.tolist(),
                                     [s.epoch for s in samp])
            return finished

        use_streams = p_batch is not None and self.runner.is_gpu
        if use_streams:
            pre_ev = torch.cuda.Event()
            pre_ev.record()

        pend = self._pending
        sampled = pend.tokens if pend is not None else None
        pending_map = (pend.tokens, pend.index) if pend is not None else None
        logits = self.runner.execute(d_batch, sampled, pending_map)
        tokens = self.sampler.sample(logits, d_batch.seqs)
        for seq in d_batch.seqs:
            seq.sched_len = seq.sched_tokens + 1
        seqs_all = list(d_batch.seqs)
        parts = [tokens]

        if p_batch is not None:
            if use_streams:
                side = self._side_stream()
                side.wait_event(pre_ev)
                with torch.cuda.stream(side):
                    hidden = self.runner.execute_prefill(p_batch.chunks)
                    samp_pre = p_batch.sampling_seqs
                    tok_p = self._sample_maybe_pp(hidden, samp_pre) \
                        if samp_pre else None
                if tok_p is not None:
                    ev = torch.cuda.Event()
                    ev.record(side)
                    torch.cuda.current_stream().wait_event(ev)
                    tok_p.record_stream(torch.cuda.current_stream())
            else:
                hidden = self.runner.execute_prefill(p_batch.chunks)
                samp_pre = p_batch.sampling_seqs
                tok_p = self._sample_maybe_pp(hidden, samp_pre) \
                    if samp_pre else None
            samp = self._prefill_bookkeep(p_batch)
            if tok_p is not None:
                seqs_all += samp
                parts.append(tok_p)

        tokens = torch.cat(parts) if len(parts) > 1 else parts[0]
        host_copy = None
        if tokens.is_cuda:
            host_copy = self._pinned(len(tokens))
            host_copy.copy_(tokens, non_blocking=True)
            event = torch.cuda.Event()
            event.record()
        else:
            event = None
        new_pend = _PendingStep(
            seqs=seqs_all,
            epochs=[s.epoch for s in seqs_all],
            tokens=tokens,
            host=host_copy,
            event=event,
            index={s.seq_id: i for i, s in enumerate(seqs_all)})
        finished += self._resolve_pending()
        self._pending = new_pend
        return finished

    @torch.no_grad()
    def _step_classic(self) -> List[Sequence]:
        """Either/or stepping (PP lockstep, or enable_mixed_batch=False)."""
        batch = self.scheduler.schedule()
        if batch is None:
            return self._resolve_pending()
        finished: List[Sequence] = []
        if batch.is_prefill:
            # prefill steps are synchronous: drain the pipeline first so the
            # prefill batch sees fully-committed state.
            finished += self._resolve_pending()
            hidden = self.runner.execute_prefill(batch.chunks)
            samp = self._prefill_bookkeep(batch)
            if not samp:
                return finished          # all chunks partial: no sampling
            tokens = self._sample_maybe_pp(hidden, samp)
            finished += self._commit(samp, tokens.tolist(),
                                     [s.epoch for s in samp])
            return finished

        if self._pp:
            # pipeline-parallel decode runs lockstep-synchronous (no
            # speculative pipelining): every stage executes; tokens are
            # sampled on the last stage and broadcast.
            sampled = getattr(self, "_pp_prev_tokens", None)
            logits = self.runner.execute(batch, sampled)
            tokens = self._sample_maybe_pp(
                logits, batch.seqs, precomputed_logits=True)
            self._pp_prev_tokens = tokens
            for seq in batch.seqs:
                seq.sched_len = seq.sched_tokens + 1
            finished += self._commit(batch.seqs, tokens.tolist(),
                                     [s.epoch for s in batch.seqs])
            return finished

        pend = self._pending
        sampled = pend.tokens if pend is not None else None
        pending_map = (pend.tokens, pend.index) if pend is not None else None
        logits = self.runner.execute(batch, sampled, pending_map)
        tokens = self.sampler.sample(logits, batch.seqs)
        for seq in batch.seqs:
            seq.sched_len = seq.sched_tokens + 1
        host_copy = None
        if tokens.is_cuda:
            host_copy = self._pinned(len(tokens))
            host_copy.copy_(tokens, non_blocking=True)
            event = torch.cuda.Event()
            event.record()
        else:
            event = None
        new_pend = _PendingStep(
            seqs=list(batch.seqs),
            epochs=[s.epoch for s in batch.seqs],
            tokens=tokens,
            host=host_copy,
            event=event,
            index={s.seq_id: i for i, s in enumerate(batch.seqs)})
        # resolve the PREVIOUS step while the GPU runs this one
        finished += self._resolve_pending()
        self._pending = new_pend
        return finished

    def _sample_maybe_pp(self, hidden_or_logits, seqs,
                         precomputed_logits: bool = False) -> torch.Tensor:
        """Sample on the last PP stage and broadcast token ids to every
        stage (lockstep schedulers). Single-stage: plain sampling."""
        from ..parallel import state as ps
        st = ps.get_state()
        if st.pp_size == 1:
            logits = hidden_or_logits if precomputed_logits else \
                self.runner.model.compute_logits(hidden_or_logits)
            return self.sampler.sample(logits, seqs)
        dev = self.runner.device
        if st.is_last_stage:
            logits = hidden_or_logits if precomputed_logits else \
                self.runner.model.compute_logits(hidden_or_logits)
            toks = self.sampler.sample(logits, seqs).to(dev)
        else:
            toks = torch.zeros(len(seqs), dtype=torch.long, device=dev)
        return ps.pp_broadcast_from_last(toks)

    def _pinned(self, n: int) -> torch.Tensor:
        """Pinned host staging buffers, DOUBLE-buffered: step N's async D2H
        copy must not overwrite step N-1's still-unresolved values."""
        bufs = getattr(self, "_pin_bufs", None)
        if bufs is None:
            bufs = self._pin_bufs = {}
            self._pin_flip = 0
        self._pin_flip ^= 1
        key = (n, self._pin_flip)
        t = bufs.get(key)
        if t is None:
            t = torch.empty(n, dtype=torch.long, pin_memory=True)
            bufs[key] = t
        return t

    def _resolve_pending(self) -> List[Sequence]:
        p = self._pending
        if p is None:
            return []
        self._pending = None
        if p.event is not None:
            p.event.synchronize()
            vals = p.host.tolist()
        else:
            vals = p.tokens.tolist()
        return self._commit(p.seqs, vals, p.epochs)

    def flush(self) -> List[Sequence]:
        """Drain the pipelined step (bench/end-of-stream)."""
        return self._resolve_pending()

    def _commit(self, seqs: List[Sequence], vals: List[int],
                epochs: List[int]) -> List[Sequence]:
        finished: List[Sequence] = []
        for seq, tok, ep in zip(seqs, vals, epochs):
            if seq.epoch != ep or seq.status == SeqStatus.FINISHED:
                continue  # preempted or already finished: drop stale token
            seq.append_token(int(tok))
            self.num_generation_tokens += 1
            if seq.check_finished(self.eos_token_id):
                if self.kv_offload is not None:
                    self.kv_offload.offload(
                        seq.prompt_token_ids + seq.output_token_ids,
                        seq.block_table)
                if self.kv_publisher is not None:
                    self.kv_publisher.block_removed(list(seq.block_table))
                self.scheduler.finish(seq)
                finished.append(seq)
        return finished

    # ------------------------------------------------------------- offline
    def generate(self, prompts: List[List[int]],
                 sampling: Optional[SamplingParams] = None,
                 ) -> List[Sequence]:
        ids = [self.add_request(p, sampling) for p in prompts]
        pending = set(ids)
        t0 = time.monotonic()
        while pending:
            for seq in self.step():
                pending.discard(seq.seq_id)
            if time.monotonic() - t0 > 3600:
                raise TimeoutError("generate() exceeded 1h")
        return [self.seqs[i] for i in ids]
