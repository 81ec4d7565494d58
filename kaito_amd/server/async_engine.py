"""AsyncLLMEngine: the background engine loop bridging asyncio request
handlers to the synchronous continuous-batching engine.

The engine loop owns the GPU and runs in one dedicated thread (one process
per GPU — DP replicas are separate processes behind the k8s Service). Token
streams flow back through per-request asyncio queues.
"""
from __future__ import annotations

import asyncio
import os
import queue
import threading
import time
from dataclasses import dataclass
from typing import AsyncIterator, Dict, List, Optional

from ..engine.engine import LLMEngine
from ..engine.sequence import SamplingParams
from . import metrics


@dataclass
class StreamItem:
    token_id: int
    finished: bool = False
    finish_reason: Optional[str] = None
    # [(token_id, logprob), ...] top-K + own, when requested
    logprobs: Optional[list] = None



@dataclass
class _Pending:
    prompt: List[int]
    sampling: SamplingParams
    out_q: asyncio.Queue
    loop: asyncio.AbstractEventLoop
    seq_id: Optional[int] = None
    lora_name: Optional[str] = None


class AsyncLLMEngine:
    def __init__(self, engine: LLMEngine):
        self.engine = engine
        self._submit: "queue.Queue[_Pending]" = queue.Queue()
        self._streams: Dict[int, _Pending] = {}
        self._emitted: Dict[int, int] = {}
        self._wake = threading.Event()
        self._stop = False
        self._thread: Optional[threading.Thread] = None

    # ---------------------------------------------------------------- loop
    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="kaito-engine-loop")
        self._thread.start()
        return self

    def shutdown(self):
        self._stop = True
        self._wake.set()
        if self._thread:
            self._thread.join(timeout=10)

    def _drain_submissions(self):
        while True:
            try:
                p = self._submit.get_nowait()
            except queue.Empty:
                return
            try:
                sid = self.engine.add_request(p.prompt, p.sampling,
                                              lora_name=p.lora_name)
            except Exception as e:  # noqa: BLE001 — bad request (unknown
                # LoRA, over-long prompt from a non-HTTP caller): fail ONLY
                # this stream; a raise here would kill the engine thread
                # and hang every in-flight and future request
                self._push(p, StreamItem(-1, True, f"error: {e}"))
                continue
            p.seq_id = sid
            self._streams[sid] = p
            self._emitted[sid] = 0

    def _fail_all(self, reason: str):
        """Engine-loop exception: fail every in-flight stream (the loop
        must survive bad requests — a dead loop hangs all clients)."""
        for sid, p in list(self._streams.items()):
            self._push(p, StreamItem(-1, True, reason))
            self.engine.abort(sid)
            self._streams.pop(sid, None)
            self._emitted.pop(sid, None)

    def _loop(self):
        eng = self.engine
        hb = None
        if os.environ.get("KAITO_HEARTBEAT", "1") != "0":
            from .heartbeat import Heartbeat
            hb = Heartbeat(int(os.environ.get("RANK", "0")))
        while not self._stop:
            if hb is not None:
                hb.beat()   # per-rank liveness (multi_node_health_check)
            self._drain_submissions()
            if not eng.has_unfinished():
                metrics.REQUESTS_RUNNING.set(0)
                self._wake.wait(timeout=0.05)
                self._wake.clear()
                continue
            try:
                finished = eng.step()
            except Exception:  # noqa: BLE001
                import logging
                logging.getLogger("kaito_amd.server").exception(
                    "engine step failed; failing in-flight requests")
                self._fail_all("error")
                continue
            metrics.REQUESTS_RUNNING.set(eng.scheduler.num_running)
            metrics.REQUESTS_WAITING.set(eng.scheduler.num_waiting)
            hit = getattr(eng.pool, "hit_tokens", None)
            if hit is not None:
                metrics.PREFIX_CACHE_HIT_TOKENS.set(hit)
            nb = eng.pool.num_blocks
            if nb:
                metrics.GPU_CACHE_USAGE.set(1.0 - eng.pool.num_free / nb)
            # push fresh tokens to streams
            for sid, p in list(self._streams.items()):
                seq = eng.seqs.get(sid)
                if seq is None:
                    continue
                n = len(seq.output_token_ids)
                e = self._emitted[sid]
                if n > e:
                    metrics.GENERATION_TOKENS.inc(n - e)
                    lps = seq.output_logprobs
                    for j in range(e, n):
                        it = StreamItem(seq.output_token_ids[j])
                        if j < len(lps):
                            it.logprobs = lps[j]
                        self._push(p, it)
                    self._emitted[sid] = n
            for seq in finished:
                p = self._streams.pop(seq.seq_id, None)
                self._emitted.pop(seq.seq_id, None)
                if p is not None:
                    metrics.PROMPT_TOKENS.inc(seq.num_prompt_tokens)
                    fin = seq.finish_time or time.monotonic()
                    metrics.E2E_LATENCY.inc(fin - seq.arrival_time)
                    if seq.first_token_time is not None:
                        metrics.TTFT.observe(
                            seq.first_token_time - seq.arrival_time)
                        n_out = len(seq.output_token_ids)
                        if n_out > 1:
                            metrics.TPOT.observe(
                                (fin - seq.first_token_time) / (n_out - 1))
                    self._push(p, StreamItem(-1, True, seq.finish_reason))

    @staticmethod
    def _push(p: _Pending, item: StreamItem):
        p.loop.call_soon_threadsafe(p.out_q.put_nowait, item)

    # ---------------------------------------------------------------- API
    @property
    def num_waiting(self) -> int:
        return self.engine.scheduler.num_waiting + self._submit.qsize()

    @property
    def num_running(self) -> int:
        return self.engine.scheduler.num_running

    def abort(self, seq_id: Optional[int]) -> None:
        """Client-side cancel (stop-string hit / disconnect)."""
        if seq_id is None:
            return
        self._streams.pop(seq_id, None)
        self._emitted.pop(seq_id, None)
        self.engine.abort(seq_id)

    async def generate(self, prompt_ids: List[int], sampling: SamplingParams,
                       lora_name: Optional[str] = None
                       ) -> AsyncIterator[StreamItem]:
        loop = asyncio.get_running_loop()
        out_q: asyncio.Queue = asyncio.Queue()
        p = _Pending(prompt_ids, sampling, out_q, loop, lora_name=lora_name)
        self._submit.put(p)
        self._wake.set()
        try:
            while True:
                item = await out_q.get()
                yield item
                if item.finished:
                    return
        finally:
            # consumer stopped early (stop string / client disconnect)
            self.abort(p.seq_id)
