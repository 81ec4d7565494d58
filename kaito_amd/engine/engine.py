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
    # logprob parts: (host_vals [n,K+1], host_ids [n,K+1], start_row) per
    # sampling call that produced logprobs (decode part / prefill part)
    lp_parts: List[tuple] = None


@dataclass
class _PendingPrefill:
    """A fire-and-forget prefill chunk in flight on the side stream."""
    seqs: List[Sequence]                 # sampling (prompt-completing) seqs
    epochs: List[int]
    tokens: Optional[torch.Tensor]       # first sampled tokens (device)
    host: Optional[torch.Tensor]         # pinned copy (GPU) or None
    event: Optional[object]              # side-stream cuda Event or None
    lp_part: Optional[tuple]             # (host_vals, host_ids) or None


class LLMEngine:
    def __init__(self, cfg: EngineConfig, weights_path: Optional[str] = None):
        self.cfg = cfg
        if cfg.model.is_mla:
            # MLA (deepseek): chunked prefill + prefix caching work over
            # the latent cache (MLAAttention._context), but host offload
            # stays off — the aliased (c, c) cache pair would double the
            # offloaded bytes in KVOffloadManager (models/mla.py)
            cfg.kv_offload = False
        self._maybe_enable_one_shot(cfg)
        self.runner = ModelRunner(cfg).load_model(weights_path, cfg.seed)
        self.runner.setup_tunable()
        self.runner.profile_and_allocate_kv()
        if cfg.enable_prefix_caching:
            from .block_pool import PrefixCachingPool
            self.pool = PrefixCachingPool(self.runner.num_gpu_blocks,
                                          cfg.block_size)
        else:
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
        self._prefill_pend: Optional[_PendingPrefill] = None
        # counters for /metrics (names consumed by the benchmark probe)
        self.num_generation_tokens = 0
        self.num_prompt_tokens = 0
        # optional KV event bus (EPP KVCache-aware routing surface)
        self.kv_publisher = None

    @staticmethod
    def _maybe_enable_one_shot(cfg: EngineConfig) -> None:
        """Activate the fused one-shot allreduce+RMSNorm group for TP
        decode (xGMI hipIpc peer staging; gloo-emulated on CPU). pp>1 is
        excluded: the stage-boundary residual fold has no norm call site
        to absorb a deferred reduce."""
        from ..parallel.state import get_state
        st = get_state()
        want = cfg.enable_one_shot_allreduce
        if want is None:
            want = st.tp_size > 1 and st.pp_size == 1
        if not (want and st.tp_size > 1 and st.pp_size == 1):
            return
        from ..parallel import one_shot
        if one_shot.active() is not None:
            return
        one_shot.activate(one_shot.make_group(
            cfg.max_num_seqs, cfg.model.hidden_size, cfg.model.dtype))

    def capture_graphs(self):
        self.runner.capture_decode_graphs()
        return self

    # ------------------------------------------------------------- requests
    def add_request(self, prompt_token_ids: List[int],
                    sampling: Optional[SamplingParams] = None,
                    seq_id: Optional[int] = None,
                    lora_name: Optional[str] = None) -> int:
        sid = seq_id if seq_id is not None else next(self._next_id)
        sampling = sampling or SamplingParams()
        # clamp generation to the model context: an unclamped request
        # would outgrow the fixed per-seq block-table width
        room = self.runner.max_model_len - len(prompt_token_ids)
        if room <= 0:
            raise ValueError(
                f"prompt ({len(prompt_token_ids)} tokens) leaves no room "
                f"in max_model_len={self.runner.max_model_len}")
        if sampling.max_tokens > room:
            import dataclasses
            sampling = dataclasses.replace(sampling, max_tokens=room)
        seq = Sequence(sid, list(prompt_token_ids), sampling)
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
        pp = self._prefill_pend
        if pp is not None and any(s2.seq_id == seq_id for s2 in pp.seqs) \
            and pp.event is not None:
            pp.event.synchronize()   # rare: in-flight prefill being aborted
        if seq.status == SeqStatus.RUNNING:
            self.scheduler.finish(seq)
        elif seq in self.scheduler.waiting:
            self.scheduler.waiting.remove(seq)
        seq.status = SeqStatus.FINISHED
        seq.finish_reason = "abort"

    def has_unfinished(self) -> bool:
        return self.scheduler.has_work() or self._pending is not None \
            or self._prefill_pend is not None

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
            seq.sched_len = seq.num_context_tokens + 1
            self.num_prompt_tokens += seq.num_prompt_tokens
            self._register_prefix(seq)
        if self.kv_publisher is not None and samp:
            blocks = [b for s2 in samp for b in s2.block_table]
            self.kv_publisher.block_stored(blocks)
        return samp

    def _sample_lp(self, logits, seqs):
        """sampler.sample with logprob staging: returns (tokens, part) where
        part = (host_vals, host_ids) copied async, or None."""
        tokens, lpv, lpi = self.sampler.sample(logits, seqs,
                                               return_logprobs=True)
        part = None
        if lpv is not None:
            if lpv.is_cuda:
                hv = torch.empty_like(lpv, device="cpu", pin_memory=True)
                hi = torch.empty_like(lpi, device="cpu", pin_memory=True)
                hv.copy_(lpv, non_blocking=True)
                hi.copy_(lpi, non_blocking=True)
                part = (hv, hi)
            else:
                part = (lpv, lpi)
        return tokens, part

    @staticmethod
    def _lp_rows(parts, n):
        """Expand pending lp_parts into one per-row list (None where the
        row's step produced no logprobs)."""
        rows = [None] * n
        for hv, hi, start in parts or []:
            v = hv.tolist()
            i = hi.tolist()
            for r in range(len(v)):
                rows[start + r] = list(zip(i[r], v[r]))
        return rows

    def _register_prefix(self, seq) -> None:
        """Publish a completed prompt's full blocks to the prefix cache."""
        if hasattr(self.pool, "register_prefix") and seq.block_table:
            bs = self.cfg.block_size
            nfull = seq.num_prompt_tokens // bs
            self.pool.register_prefix(seq.prompt_token_ids[:nfull * bs],
                                      seq.block_table[:nfull])

    def _side_stream(self):
        s = getattr(self, "_side", None)
        if s is None:
            s = self._side = torch.cuda.Stream()
        return s

    @torch.no_grad()
    def _step_mixed(self) -> List[Sequence]:
        """Mixed step: decode over all running seqs on the main stream
        (hipGraph), with a bounded prefill chunk launched FIRE-AND-FORGET
        on a side stream. Prefill's MFMA-bound GEMMs overlap decode's
        bandwidth-bound paged attention; the decode chain NEVER waits on
        the side stream — prefill results resolve via a non-blocking
        event.query() on a later step, and a prefilled sequence only joins
        the decode set after its event fired (its KV writes are complete).

        Ordering: the side stream waits on an event recorded BEFORE this
        step's decode launch. That event orders the prefill after (a) the
        previous step's decode KV writes — whose target blocks may have
        been freed by a finish/preemption and re-allocated to a prefill
        seq this step — and (b) this step's prefix-restore H2D copies,
        without serializing it after this step's decode."""
        finished: List[Sequence] = []
        finished += self._resolve_prefill(block=False)
        allow_prefill = self._prefill_pend is None
        d_batch, p_batch = self.scheduler.schedule_mixed(
            self.cfg.mixed_prefill_tokens, allow_prefill)
        if d_batch is None and p_batch is None:
            if self._prefill_pend is not None:
                finished += self._resolve_prefill(block=True)
                return finished          # next step schedules the promotees
            return finished + self._resolve_pending()
        if d_batch is None:
            # startup burst: classic synchronous full-budget prefill
            finished += self._resolve_prefill(block=True)
            finished += self._resolve_pending()
            hidden = self.runner.execute_prefill(p_batch.chunks)
            samp = self._prefill_bookkeep(p_batch)
            if not samp:
                return finished
            tokens, lp = self._sample_prefill_pp(hidden, samp)
            finished += self._commit(samp, tokens.tolist(),
                                     [s.epoch for s in samp],
                                     self._lp_rows([lp + (0,)], len(samp))
                                     if lp is not None else None)
            return finished

        use_streams = p_batch is not None and self.runner.is_gpu
        if use_streams:
            pre_ev = torch.cuda.Event()
            pre_ev.record()

        if any(s.sampling.needs_history for s in d_batch.seqs):
            # penalties need exact token history: drain the pipeline
            finished += self._resolve_pending()
        pend = self._pending
        sampled = pend.tokens if pend is not None else None
        pending_map = (pend.tokens, pend.index) if pend is not None else None
        logits = self.runner.execute(d_batch, sampled, pending_map)
        tokens, lp_d = self._sample_lp(logits, d_batch.seqs)
        for seq in d_batch.seqs:
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
            seqs=list(d_batch.seqs),
            epochs=[s.epoch for s in d_batch.seqs],
            tokens=tokens,
            host=host_copy,
            event=event,
            index={s.seq_id: i for i, s in enumerate(d_batch.seqs)},
            lp_parts=[(lp_d[0], lp_d[1], 0)] if lp_d is not None else [])

        if p_batch is not None:
            self._launch_prefill(p_batch, pre_ev if use_streams else None)

        finished += self._resolve_pending()
        self._pending = new_pend
        return finished

    def _launch_prefill(self, p_batch, pre_ev) -> None:
        """Fire-and-forget prefill chunk on the side stream."""
        samp_pre = p_batch.sampling_seqs
        if pre_ev is not None:
            side = self._side_stream()
            side.wait_event(pre_ev)
            with torch.cuda.stream(side):
                hidden = self.runner.execute_prefill(p_batch.chunks)
                tok_p = lp_p = host = None
                if samp_pre:
                    logits_p = self.runner.model.compute_logits(hidden)
                    tok_p, lp_p = self._sample_lp(logits_p, samp_pre)
                    host = torch.empty(len(samp_pre), dtype=torch.long,
                                       pin_memory=True)
                    host.copy_(tok_p, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record(side)
        else:
            hidden = self.runner.execute_prefill(p_batch.chunks)
            tok_p = lp_p = host = None
            ev = None
            if samp_pre:
                logits_p = self.runner.model.compute_logits(hidden)
                tok_p, lp_p = self._sample_lp(logits_p, samp_pre)
        # bookkeeping at launch: advance chunks, count prompt tokens;
        # PROMOTION to the decode set waits for the event (resolve)
        self.scheduler.advance_prefill_chunks(p_batch)
        for seq in samp_pre:
            seq.sched_len = seq.num_context_tokens + 1
            self.num_prompt_tokens += seq.num_prompt_tokens
        if self.kv_publisher is not None and samp_pre:
            blocks = [b for s2 in samp_pre for b in s2.block_table]
            self.kv_publisher.block_stored(blocks)
        self._prefill_pend = _PendingPrefill(
            seqs=list(samp_pre),
            epochs=[s.epoch for s in samp_pre],
            tokens=tok_p, host=host, event=ev,
            lp_part=lp_p)

    def _resolve_prefill(self, block: bool) -> List[Sequence]:
        """Resolve the in-flight prefill chunk if its event fired (or
        always, when block=True): commit first tokens and promote the
        sequences into the decode set."""
        p = self._prefill_pend
        if p is None:
            return []
        if p.event is not None:
            if not block and not p.event.query():
                return []
            p.event.synchronize()
            vals = p.host.tolist() if p.host is not None else []
        else:
            vals = p.tokens.tolist() if p.tokens is not None else []
        self._prefill_pend = None
        lps = self._lp_rows([p.lp_part + (0,)], len(p.seqs)) \
            if p.lp_part is not None else None
        out = self._commit(p.seqs, vals, p.epochs, lps)
        self.scheduler.promote_prefilled(p.seqs)
        for seq in p.seqs:
            self._register_prefix(seq)
        return out

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
            tokens, lp = self._sample_prefill_pp(hidden, samp)
            finished += self._commit(samp, tokens.tolist(),
                                     [s.epoch for s in samp],
                                     self._lp_rows([lp + (0,)], len(samp))
                                     if lp is not None else None)
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

        if any(s.sampling.needs_history for s in batch.seqs):
            finished += self._resolve_pending()
        pend = self._pending
        sampled = pend.tokens if pend is not None else None
        pending_map = (pend.tokens, pend.index) if pend is not None else None
        logits = self.runner.execute(batch, sampled, pending_map)
        tokens, lp_d = self._sample_lp(logits, batch.seqs)
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
            index={s.seq_id: i for i, s in enumerate(batch.seqs)},
            lp_parts=[(lp_d[0], lp_d[1], 0)] if lp_d is not None else [])
        # resolve the PREVIOUS step while the GPU runs this one
        finished += self._resolve_pending()
        self._pending = new_pend
        return finished

    def _sample_prefill_pp(self, hidden, seqs):
        """Prefill-completion sampling: logprobs on single-stage; PP
        lockstep broadcasts token ids only (logprobs unsupported under
        PP)."""
        from ..parallel import state as ps
        st = ps.get_state()
        if st.pp_size == 1:
            logits = self.runner.model.compute_logits(hidden)
            return self._sample_lp(logits, seqs)
        return self._sample_maybe_pp(hidden, seqs), None

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
        lps = self._lp_rows(p.lp_parts, len(p.seqs)) if p.lp_parts else None
        return self._commit(p.seqs, vals, p.epochs, lps)

    def flush(self) -> List[Sequence]:
        """Drain the pipelined step + in-flight prefill."""
        return self._resolve_prefill(block=True) + self._resolve_pending()

    def _commit(self, seqs: List[Sequence], vals: List[int],
                epochs: List[int],
                lps: Optional[List] = None) -> List[Sequence]:
        finished: List[Sequence] = []
        for i, (seq, tok, ep) in enumerate(zip(seqs, vals, epochs)):
            if seq.epoch != ep or seq.status == SeqStatus.FINISHED:
                continue  # preempted or already finished: drop stale token
            seq.append_token(int(tok))
            k = seq.sampling.logprobs
            if k and lps is not None and lps[i] is not None:
                # row = top-K(batch max) pairs + the sampled token's own
                # logprob as the last entry; trim to this seq's K
                row = lps[i]
                seq.output_logprobs.append(row[:k] + [row[-1]])
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
