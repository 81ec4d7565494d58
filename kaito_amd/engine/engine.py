"""LLMEngine: the continuous-batching serving loop.

The MI355X-native replacement for the vLLM engine the reference's Workspace
pods launch (SURVEY.md §3.2 "HOT LOOP"). One engine per GPU (DP tier) or per
TP group; step() = schedule → forward → sample → append/finish.
"""
from __future__ import annotations

import itertools
import logging
import time
from typing import Dict, List, Optional

import torch

from .config import EngineConfig
from .block_pool import BlockPool
from .model_runner import ModelRunner
from .sampler import Sampler
from .scheduler import Scheduler
from .sequence import SamplingParams, Sequence, SeqStatus

logger = logging.getLogger(__name__)


class LLMEngine:
    def __init__(self, cfg: EngineConfig, weights_path: Optional[str] = None):
        self.cfg = cfg
        self.runner = ModelRunner(cfg).load_model(weights_path, cfg.seed)
        self.runner.setup_tunable()
        self.runner.profile_and_allocate_kv()
        self.pool = BlockPool(self.runner.num_gpu_blocks, cfg.block_size)
        self.scheduler = Scheduler(cfg, self.pool)
        self.sampler = Sampler(cfg.device, cfg.seed)
        self.eos_token_id: Optional[int] = None
        self._next_id = itertools.count()
        self.seqs: Dict[int, Sequence] = {}
        # counters for /metrics (names consumed by the benchmark probe)
        self.num_generation_tokens = 0
        self.num_prompt_tokens = 0

    def capture_graphs(self):
        self.runner.capture_decode_graphs()
        return self

    # ------------------------------------------------------------- requests
    def add_request(self, prompt_token_ids: List[int],
                    sampling: Optional[SamplingParams] = None,
                    seq_id: Optional[int] = None) -> int:
        sid = seq_id if seq_id is not None else next(self._next_id)
        seq = Sequence(sid, list(prompt_token_ids),
                       sampling or SamplingParams())
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
        return self.scheduler.has_work()

    # ------------------------------------------------------------- stepping
    @torch.no_grad()
    def step(self) -> List[Sequence]:
        """One engine iteration. Returns sequences that FINISHED this step."""
        batch = self.scheduler.schedule()
        if batch is None:
            return []
        sampled = None if batch.is_prefill else getattr(self, "_last_decode_tokens", None)
        logits = self.runner.execute(batch, sampled)
        tokens = self.sampler.sample(logits, batch.seqs)
        if not batch.is_prefill:
            self._last_decode_tokens = tokens
        tokens_cpu = tokens.tolist()
        finished: List[Sequence] = []
        for seq, tok in zip(batch.seqs, tokens_cpu):
            seq.append_token(int(tok))
            self.num_generation_tokens += 1
            if batch.is_prefill:
                self.num_prompt_tokens += seq.num_prompt_tokens
            if seq.check_finished(self.eos_token_id):
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
