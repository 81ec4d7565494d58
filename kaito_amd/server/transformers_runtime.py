"""Fallback serving runtime — parity with the reference's transformers
preset (presets/workspace/inference/text-generation/inference_api.py:
OpenAI-compatible serving via HF transformers for models outside the
native engine's support matrix; selected by the kaito.sh/runtime
annotation).

Backends:
  * HF transformers AutoModelForCausalLM when --weights-path is given
    (torch-ROCm; GEMMs on MFMA via hipBLASLt)
  * registry model with random-init weights otherwise (air-gapped smoke),
    decoded by full-recompute through the prefill path — slow but covers
    every architecture the model classes express (e.g. head_dim=96).

Requests are served one at a time (no continuous batching — that is the
native engine's job; this runtime is the compatibility fallback).
"""
from __future__ import annotations

import argparse
import asyncio
import os
import time
import uuid
from typing import List, Optional

import torch
from fastapi import FastAPI
from fastapi.responses import Response

from ..engine.config import ModelConfig
from . import metrics
from .tokenizer import load_tokenizer


class FallbackGenerator:
    def __init__(self, model_name: str, weights_path: Optional[str] = None,
                 device: Optional[str] = None):
        self.device = device or (
            "cuda" if torch.cuda.is_available() else "cpu")
        self.hf = None
        self.tokenizer = None
        if weights_path:
            from transformers import AutoModelForCausalLM
            self.hf = AutoModelForCausalLM.from_pretrained(
                weights_path,
                torch_dtype=torch.bfloat16 if self.device == "cuda"
                else torch.float32).to(self.device).eval()
            self.tokenizer = load_tokenizer(weights_path)
            self.vocab_size = self.hf.config.vocab_size
        else:
            from ..models import get_model_config
            from ..models.llama import LlamaForCausalLM
            from ..parallel.state import init_parallel
            init_parallel(1)
            cfg: ModelConfig = get_model_config(model_name)
            self.cfg = cfg
            self.model = LlamaForCausalLM(cfg).to(self.device)
            self.model.random_init(0)
            self.model.init_rope(self.device, cfg.max_position)
            self.model.eval()
            self.tokenizer = load_tokenizer(None, cfg.vocab_size)
            self.vocab_size = cfg.vocab_size
        self._lock = asyncio.Lock()

    @torch.no_grad()
    def _generate_native(self, prompt: List[int], max_tokens: int,
                         temperature: float) -> List[int]:
        from ..models.llama import AttnMetadata
        toks = list(prompt)
        out: List[int] = []
        for _ in range(max_tokens):
            T = len(toks)
            meta = AttnMetadata(
                is_prefill=True,
                slot_mapping=torch.full((T,), -1, dtype=torch.long,
                                        device=self.device),
                cu_seqlens=torch.tensor([0, T], dtype=torch.int32,
                                        device=self.device),
                max_seqlen=T)
            hidden = self.model(
                torch.tensor(toks, device=self.device),
                torch.arange(T, device=self.device), None, meta)
            logits = self.model.compute_logits(hidden[-1:]).float()
            if temperature > 0:
                probs = torch.softmax(logits / temperature, -1)
                nxt = int(torch.multinomial(probs, 1))
            else:
                nxt = int(logits.argmax(-1))
            toks.append(nxt)
            out.append(nxt)
        return out

    @torch.no_grad()
    def _generate_hf(self, prompt: List[int], max_tokens: int,
                     temperature: float) -> List[int]:
        ids = torch.tensor([prompt], device=self.device)
        gen = self.hf.generate(
            ids, max_new_tokens=max_tokens,
            do_sample=temperature > 0,
            temperature=max(temperature, 1e-5) if temperature > 0 else None,
            pad_token_id=getattr(self.tokenizer, "pad_token_id", 0) or 0)
        return gen[0, ids.shape[1]:].tolist()

    async def generate(self, prompt: List[int], max_tokens: int,
                       temperature: float) -> List[int]:
        async with self._lock:                 # one request at a time
            fn = self._generate_hf if self.hf is not None \
                else self._generate_native
            return await asyncio.get_running_loop().run_in_executor(
                None, fn, prompt, max_tokens, temperature)


def build_fallback_app(gen: FallbackGenerator, model_name: str) -> FastAPI:
    app = FastAPI(title="kaito-amd transformers fallback runtime")

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/metrics")
    async def prom():
        data, ctype = metrics.render()
        return Response(content=data, media_type=ctype)

    @app.get("/v1/models")
    async def models():
        return {"object": "list", "data": [{
            "id": model_name, "object": "model", "owned_by": "kaito-amd",
            "vocab_size": gen.vocab_size, "runtime": "transformers"}]}

    @app.post("/v1/completions")
    async def completions(body: dict):
        prompt = body.get("prompt", "")
        if isinstance(prompt, list) and prompt and isinstance(prompt[0], int):
            ids = [int(x) for x in prompt]
        else:
            ids = gen.tokenizer.encode(str(prompt))
        toks = await gen.generate(ids, int(body.get("max_tokens", 16)),
                                  float(body.get("temperature", 1.0)))
        metrics.GENERATION_TOKENS.inc(len(toks))
        return {"id": f"cmpl-{uuid.uuid4().hex[:24]}",
                "object": "text_completion", "created": int(time.time()),
                "model": model_name,
                "choices": [{"index": 0, "text": gen.tokenizer.decode(toks),
                             "finish_reason": "length"}],
                "usage": {"prompt_tokens": len(ids),
                          "completion_tokens": len(toks),
                          "total_tokens": len(ids) + len(toks)}}

    @app.post("/v1/chat/completions")
    async def chat(body: dict):
        text = gen.tokenizer.apply_chat_template(
            body.get("messages", []), add_generation_prompt=True,
            tokenize=False)
        ids = gen.tokenizer.encode(text)
        toks = await gen.generate(
            ids, int(body.get("max_tokens") or 128),
            float(body.get("temperature", 1.0)))
        metrics.GENERATION_TOKENS.inc(len(toks))
        return {"id": f"chatcmpl-{uuid.uuid4().hex[:24]}",
                "object": "chat.completion", "created": int(time.time()),
                "model": model_name,
                "choices": [{"index": 0,
                             "message": {"role": "assistant",
                                         "content": gen.tokenizer.decode(toks)},
                             "finish_reason": "stop"}]}

    return app


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--model", default=os.environ.get("KAITO_MODEL", ""))
    p.add_argument("--weights-path", default=os.environ.get("KAITO_WEIGHTS_PATH"))
    p.add_argument("--port", type=int, default=5000)
    p.add_argument("--host", default="0.0.0.0")
    args = p.parse_args(argv)
    gen = FallbackGenerator(args.model, args.weights_path)
    app = build_fallback_app(gen, args.model)
    import uvicorn
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
