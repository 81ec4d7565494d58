"""Request/sequence state for the continuous-batching engine."""
from __future__ import annotations

import enum
import time
from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class SamplingParams:
    max_tokens: int = 128
    temperature: float = 0.0          # 0 → greedy
    top_p: float = 1.0
    top_k: int = 0                    # 0 → disabled
    stop_token_ids: tuple = ()
    ignore_eos: bool = False
    seed: Optional[int] = None
    # OpenAI-style penalties over GENERATED tokens (presence/frequency) or
    # prompt+generated (repetition). Using any forces synchronous sampling
    # (exact token history; the pipelined step is drained first).
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0
    # top-N logprobs returned per sampled token (None = off)
    logprobs: Optional[int] = None

    @property
    def needs_history(self) -> bool:
        return (self.presence_penalty != 0.0 or self.frequency_penalty != 0.0
                or self.repetition_penalty != 1.0)


class SeqStatus(enum.Enum):
    WAITING = 0
    RUNNING = 1
    FINISHED = 2


@dataclass
class Sequence:
    seq_id: int
    prompt_token_ids: List[int]
    sampling: SamplingParams = field(default_factory=SamplingParams)
    status: SeqStatus = SeqStatus.WAITING
    output_token_ids: List[int] = field(default_factory=list)
    block_table: List[int] = field(default_factory=list)
    arrival_time: float = field(default_factory=time.monotonic)
    first_token_time: Optional[float] = None
    finish_time: Optional[float] = None
    finish_reason: Optional[str] = None

    # tokens scheduled on-device but not yet resolved to host (pipelined
    # decode); sched_tokens is the authoritative length for KV/blocks.
    sched_len: int = 0
    # bumped on preemption so stale in-flight results are dropped at resolve
    epoch: int = 0
    lora_id: int = -1            # adapter slot (-1 = base model)
    # prompt tokens whose KV is in the cache (chunked prefill / prefix
    # restore); prompt fully prefilled when == num_prompt_tokens
    prefilled_len: int = 0
    # per output token: [(token_id, logprob), ...] top-N, when
    # sampling.logprobs is set
    output_logprobs: List[list] = field(default_factory=list)

    @property
    def num_prompt_tokens(self) -> int:
        return len(self.prompt_token_ids)

    @property
    def context_token_ids(self) -> List[int]:
        """Tokens whose KV must be in the cache before the next decode:
        prompt + already-generated output. After a recompute preemption the
        generated tokens are preserved and re-prefilled as context (vLLM
        recompute semantics), so streamed clients never see a divergent
        re-generation of indices they already received."""
        if not self.output_token_ids:
            return self.prompt_token_ids
        return self.prompt_token_ids + self.output_token_ids

    @property
    def num_context_tokens(self) -> int:
        return len(self.prompt_token_ids) + len(self.output_token_ids)

    @property
    def num_tokens(self) -> int:
        return len(self.prompt_token_ids) + len(self.output_token_ids)

    @property
    def sched_tokens(self) -> int:
        return max(self.num_tokens, self.sched_len)

    @property
    def sched_output_tokens(self) -> int:
        return self.sched_tokens - len(self.prompt_token_ids)

    @property
    def last_token_id(self) -> int:
        return (self.output_token_ids[-1] if self.output_token_ids
                else self.prompt_token_ids[-1])

    def append_token(self, tok: int) -> None:
        if self.first_token_time is None:
            self.first_token_time = time.monotonic()
        self.output_token_ids.append(tok)

    def check_finished(self, eos_token_id: Optional[int]) -> bool:
        sp = self.sampling
        if len(self.output_token_ids) >= sp.max_tokens:
            self.finish_reason = "length"
        elif not sp.ignore_eos and self.output_token_ids:
            last = self.output_token_ids[-1]
            if (eos_token_id is not None and last == eos_token_id) or \
               last in sp.stop_token_ids:
                self.finish_reason = "stop"
        if self.finish_reason:
            self.status = SeqStatus.FINISHED
            self.finish_time = time.monotonic()
            return True
        return False
