"""OpenAI-compatible HTTP front-end on :5000.

Endpoint surface matches what the reference's ecosystem expects from a
Workspace pod: /health, /metrics, /v1/models,
/v1/completions, /v1/chat/completions (with SSE streaming). Metric names
are vLLM-compatible so benchmark_entrypoint / EPP / KEDA integrations work
unchanged.
"""
from __future__ import annotations

import asyncio
import dataclasses
import json
import time
import uuid
from typing import AsyncIterator, List, Optional

from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import Response, StreamingResponse
from pydantic import BaseModel, Field

from ..engine.sequence import SamplingParams
from . import metrics
from .async_engine import AsyncLLMEngine
from .rate_limit import RateLimitMiddleware


class CompletionRequest(BaseModel):
    model: str = ""
    prompt: object = ""           # str | list[str] | list[int]
    max_tokens: int = 16
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = 0
    n: int = 1
    stream: bool = False
    stop: Optional[object] = None
    ignore_eos: bool = False
    seed: Optional[int] = None
    logprobs: Optional[int] = None
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0


class ChatMessage(BaseModel):
    role: str
    content: str


class ChatCompletionRequest(BaseModel):
    model: str = ""
    messages: List[ChatMessage] = Field(default_factory=list)
    max_tokens: Optional[int] = None
    max_completion_tokens: Optional[int] = None
    temperature: float = 1.0
    top_p: float = 1.0
    stream: bool = False
    ignore_eos: bool = False
    logprobs: bool = False
    top_logprobs: Optional[int] = None
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0


def _sampling(max_tokens, temperature, top_p, top_k=0, ignore_eos=False,
              eos_id=None, logprobs=None, presence_penalty=0.0,
              frequency_penalty=0.0, repetition_penalty=1.0, seed=None):
    return SamplingParams(
        max_tokens=max_tokens or 16,
        temperature=temperature if temperature is not None else 1.0,
        top_p=top_p, top_k=top_k, ignore_eos=ignore_eos,
        logprobs=logprobs, presence_penalty=presence_penalty,
        frequency_penalty=frequency_penalty,
        repetition_penalty=repetition_penalty, seed=seed)


def _lp_openai(lp_rows, toks, tokenizer):
    """OpenAI legacy completions logprobs object from per-token
    [(token_id, logprob), ...] rows (own-token entry last)."""
    return {
        "tokens": [tokenizer.decode([t]) for t in toks],
        "token_logprobs": [r[-1][1] if r else None for r in lp_rows],
        "top_logprobs": [
            {tokenizer.decode([tid]): v for tid, v in (r[:-1] if r else [])}
            for r in lp_rows],
        "text_offset": []}


def _stop_list(stop) -> List[str]:
    if stop is None:
        return []
    return [stop] if isinstance(stop, str) else [str(x) for x in stop]


def _truncate_at_stop(text: str, stops: List[str]):
    """Returns (text, hit): text cut before the first stop string."""
    best = None
    for st in stops:
        i = text.find(st)
        if i >= 0 and (best is None or i < best):
            best = i
    return (text[:best], True) if best is not None else (text, False)


def build_app(async_engine: AsyncLLMEngine, tokenizer, model_name: str,
              max_queue: Optional[int] = None) -> FastAPI:
    app = FastAPI(title="kaito-amd inference engine")
    eng_cfg = async_engine.engine.cfg
    vocab_size = eng_cfg.model.vocab_size
    max_len = async_engine.engine.runner.max_model_len

    def _check_prompt(ids):
        if any(i < 0 or i >= vocab_size for i in ids):
            raise HTTPException(422, f"prompt token id out of vocab "
                                     f"(vocab_size={vocab_size})")
        if len(ids) >= max_len:
            raise HTTPException(422, f"prompt longer than max_model_len "
                                     f"({max_len})")
    app.add_middleware(
        RateLimitMiddleware,
        get_queue_depth=lambda: async_engine.num_waiting,
        max_queue=max_queue or eng_cfg.max_num_seqs)

    metrics.set_cache_config(eng_cfg.block_size,
                             async_engine.engine.runner.num_gpu_blocks)

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/metrics")
    async def prom_metrics():
        data, ctype = metrics.render()
        return Response(content=data, media_type=ctype)

    @app.get("/v1/models")
    async def models():
        return {"object": "list", "data": [{
            "id": model_name, "object": "model",
            "owned_by": "kaito-amd", "created": int(time.time()),
            "vocab_size": vocab_size,
            "max_model_len": max_len}]}

    # ------------------------------------------------------------ completions
    @app.post("/v1/completions")
    async def completions(req: CompletionRequest, raw: Request):
        if isinstance(req.prompt, list) and req.prompt and \
                isinstance(req.prompt[0], int):
            prompt_ids = [int(x) for x in req.prompt]
        else:
            text = req.prompt if isinstance(req.prompt, str) else str(req.prompt)
            prompt_ids = tokenizer.encode(text)
        _check_prompt(prompt_ids)
        sp = _sampling(req.max_tokens, req.temperature, req.top_p, req.top_k,
                       req.ignore_eos, logprobs=req.logprobs,
                       presence_penalty=req.presence_penalty,
                       frequency_penalty=req.frequency_penalty,
                       repetition_penalty=req.repetition_penalty,
                       seed=req.seed)
        rid = f"cmpl-{uuid.uuid4().hex[:24]}"
        if req.stream:
            if req.n != 1:
                raise HTTPException(400, "stream with n>1 not supported")
            return StreamingResponse(
                _stream_completion(rid, prompt_ids, sp),
                media_type="text/event-stream")
        stops = _stop_list(req.stop)

        async def one_choice(index: int):
            # n>1: independent samples; the block-hash prefix cache makes
            # the shared prompt KV zero-copy across choices
            spi = sp if req.n == 1 else dataclasses.replace(
                sp, seed=(sp.seed + index if sp.seed is not None else None))
            toks: List[int] = []
            lp_rows: List[list] = []
            finish = "length"
            text = ""
            async for item in async_engine.generate(prompt_ids, spi):
                if item.finished:
                    finish = item.finish_reason or "stop"
                else:
                    toks.append(item.token_id)
                    if req.logprobs:
                        lp_rows.append(item.logprobs or [])
                    if stops:
                        text = tokenizer.decode(toks)
                        cut, hit = _truncate_at_stop(text, stops)
                        if hit:
                            text, finish = cut, "stop"
                            break
            if not stops:
                text = tokenizer.decode(toks)
            return {"index": index, "text": text, "finish_reason": finish,
                    "logprobs": _lp_openai(lp_rows, toks, tokenizer)
                    if req.logprobs else None}, len(toks)

        results = await asyncio.gather(*[one_choice(i)
                                         for i in range(max(req.n, 1))])
        choices = [r[0] for r in results]
        ntoks = sum(r[1] for r in results)
        return {
            "id": rid, "object": "text_completion",
            "created": int(time.time()), "model": model_name,
            "choices": choices,
            "usage": {"prompt_tokens": len(prompt_ids),
                      "completion_tokens": ntoks,
                      "total_tokens": len(prompt_ids) + ntoks}}

    async def _stream_completion(rid, prompt_ids, sp) -> AsyncIterator[str]:
        async for item in async_engine.generate(prompt_ids, sp):
            if item.finished:
                payload = {"id": rid, "object": "text_completion",
                           "model": model_name,
                           "choices": [{"index": 0, "text": "",
                                        "finish_reason": item.finish_reason}]}
                yield f"data: {json.dumps(payload)}\n\n"
                yield "data: [DONE]\n\n"
            else:
                payload = {"id": rid, "object": "text_completion",
                           "model": model_name,
                           "choices": [{"index": 0,
                                        "text": tokenizer.decode([item.token_id]),
                                        "finish_reason": None}]}
                yield f"data: {json.dumps(payload)}\n\n"

    # ------------------------------------------------------------ chat
    @app.post("/v1/chat/completions")
    async def chat(req: ChatCompletionRequest):
        text = tokenizer.apply_chat_template(
            [m.model_dump() for m in req.messages],
            add_generation_prompt=True, tokenize=False)
        prompt_ids = tokenizer.encode(text)
        _check_prompt(prompt_ids)
        max_toks = req.max_completion_tokens or req.max_tokens or 128
        want_lp = (req.top_logprobs or 1) if req.logprobs else None
        sp = _sampling(max_toks, req.temperature, req.top_p,
                       ignore_eos=req.ignore_eos, logprobs=want_lp,
                       presence_penalty=req.presence_penalty,
                       frequency_penalty=req.frequency_penalty)
        rid = f"chatcmpl-{uuid.uuid4().hex[:24]}"
        if req.stream:
            return StreamingResponse(_stream_chat(rid, prompt_ids, sp),
                                     media_type="text/event-stream")
        toks: List[int] = []
        lp_content: List[dict] = []
        finish = "length"
        async for item in async_engine.generate(prompt_ids, sp):
            if item.finished:
                finish = item.finish_reason or "stop"
            else:
                toks.append(item.token_id)
                if req.logprobs:
                    row = item.logprobs or []
                    lp_content.append({
                        "token": tokenizer.decode([item.token_id]),
                        "logprob": row[-1][1] if row else None,
                        "top_logprobs": [
                            {"token": tokenizer.decode([tid]), "logprob": v}
                            for tid, v in row[:-1]]})
        return {
            "id": rid, "object": "chat.completion",
            "created": int(time.time()), "model": model_name,
            "choices": [{"index": 0,
                         "message": {"role": "assistant",
                                     "content": tokenizer.decode(toks)},
                         "logprobs": {"content": lp_content}
                         if req.logprobs else None,
                         "finish_reason": finish}],
            "usage": {"prompt_tokens": len(prompt_ids),
                      "completion_tokens": len(toks),
                      "total_tokens": len(prompt_ids) + len(toks)}}

    async def _stream_chat(rid, prompt_ids, sp) -> AsyncIterator[str]:
        first = {"id": rid, "object": "chat.completion.chunk",
                 "model": model_name,
                 "choices": [{"index": 0, "delta": {"role": "assistant"},
                              "finish_reason": None}]}
        yield f"data: {json.dumps(first)}\n\n"
        async for item in async_engine.generate(prompt_ids, sp):
            if item.finished:
                payload = {"id": rid, "object": "chat.completion.chunk",
                           "model": model_name,
                           "choices": [{"index": 0, "delta": {},
                                        "finish_reason": item.finish_reason}]}
                yield f"data: {json.dumps(payload)}\n\n"
                yield "data: [DONE]\n\n"
            else:
                payload = {"id": rid, "object": "chat.completion.chunk",
                           "model": model_name,
                           "choices": [{"index": 0,
                                        "delta": {"content": tokenizer.decode([item.token_id])},
                                        "finish_reason": None}]}
                yield f"data: {json.dumps(payload)}\n\n"

    return app
