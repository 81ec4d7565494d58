"""Token sampling: greedy / temperature / top-k / top-p / penalties /
logprobs, on-device.

Operates on the logits of the batch's last tokens. Torch-native (runs on
GPU); a dedicated HIP sampling kernel is a later optimization (the sampler
is ~0 cost next to the decode forward).

Pipelining contract: all BRANCH decisions come from host-side sampling
params — sample() itself never reads device memory, so the call enqueues
async. Penalties are the exception: they need each sequence's exact token
history, so the engine drains the pipelined step before sampling when any
batch member uses them (SamplingParams.needs_history).
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from .sequence import Sequence


class Sampler:
    def __init__(self, device: str, seed: int = 0):
        self.device = device
        self.gen = None
        if device != "cpu":
            self.gen = torch.Generator(device=device)
            self.gen.manual_seed(seed)

    @staticmethod
    def _apply_penalties(logits: torch.Tensor, seqs: List[Sequence]
                         ) -> torch.Tensor:
        """OpenAI presence/frequency penalties over generated tokens and
        HF-style repetition penalty over prompt+generated. Host history →
        sparse index tensors; rows without penalties untouched."""
        out = logits
        for i, s in enumerate(seqs):
            sp = s.sampling
            if not sp.needs_history:
                continue
            if out is logits:
                out = logits.clone()
            gen_ids = s.output_token_ids
            if gen_ids and (sp.presence_penalty or sp.frequency_penalty):
                idx = torch.tensor(sorted(set(gen_ids)), dtype=torch.long,
                                   device=out.device)
                if sp.frequency_penalty:
                    counts = torch.zeros(out.shape[-1], device=out.device)
                    gi = torch.tensor(gen_ids, dtype=torch.long,
                                      device=out.device)
                    counts.scatter_add_(0, gi, torch.ones_like(
                        gi, dtype=torch.float32))
                    out[i] -= sp.frequency_penalty * counts
                if sp.presence_penalty:
                    out[i, idx] -= sp.presence_penalty
            if sp.repetition_penalty != 1.0:
                hist = set(s.prompt_token_ids) | set(gen_ids)
                idx = torch.tensor(sorted(hist), dtype=torch.long,
                                   device=out.device)
                vals = out[i, idx]
                out[i, idx] = torch.where(
                    vals > 0, vals / sp.repetition_penalty,
                    vals * sp.repetition_penalty)
        return out

    @torch.no_grad()
    def sample(self, logits: torch.Tensor, seqs: List[Sequence],
               return_logprobs: bool = False
               ) -> "torch.Tensor | Tuple[torch.Tensor, Optional[torch.Tensor], Optional[torch.Tensor]]":
        """logits: [n, vocab] (float). Returns [n] long on logits.device;
        with return_logprobs also (top_vals [n,K], top_ids [n,K]) or
        (None, None) when no seq requested logprobs."""
        if any(s.sampling.needs_history for s in seqs):
            logits = self._apply_penalties(logits, seqs)
        temps_l = [s.sampling.temperature for s in seqs]
        if all(t <= 0.0 for t in temps_l):
            tokens = logits.argmax(dim=-1)
            return self._with_logprobs(logits, tokens, seqs) \
                if return_logprobs else tokens
        temps = torch.tensor(temps_l, device=logits.device, dtype=torch.float32)
        greedy = temps <= 0.0

        logits = logits.float()
        scaled = logits / temps.clamp(min=1e-5).unsqueeze(1)

        # top-k mask (per-row k; 0 = off)
        ks = [s.sampling.top_k for s in seqs]
        if any(k > 0 for k in ks):
            for i, k in enumerate(ks):
                if k > 0:
                    kth = torch.topk(scaled[i], k).values[-1]
                    scaled[i][scaled[i] < kth] = float("-inf")

        # top-p (nucleus)
        ps_l = [s.sampling.top_p for s in seqs]
        ps = torch.tensor(ps_l, device=logits.device)
        if any(p < 1.0 for p in ps_l):
            sorted_logits, idx = torch.sort(scaled, descending=True, dim=-1)
            probs = torch.softmax(sorted_logits, dim=-1)
            cum = probs.cumsum(dim=-1)
            cut = cum - probs > ps.unsqueeze(1)   # keep first token over p
            sorted_logits[cut] = float("-inf")
            scaled = torch.full_like(scaled, float("-inf")).scatter(
                1, idx, sorted_logits)

        probs = torch.softmax(scaled, dim=-1)
        sampled = torch.multinomial(probs, 1, generator=self.gen).squeeze(1)
        # per-request seeded generators (reproducible sampling); the
        # per-row draw stays on-device so the call remains async
        for i, sq in enumerate(seqs):
            sp = sq.sampling
            if sp.seed is not None and temps_l[i] > 0.0:
                g = getattr(sq, "_sampler_gen", None)
                if g is None:
                    g = torch.Generator(device=probs.device)
                    g.manual_seed(sp.seed)
                    sq._sampler_gen = g
                sampled[i] = torch.multinomial(probs[i:i + 1], 1,
                                               generator=g)[0, 0]
        tokens = torch.where(greedy, logits.argmax(dim=-1), sampled)
        return self._with_logprobs(logits, tokens, seqs) \
            if return_logprobs else tokens

    @staticmethod
    def _with_logprobs(logits: torch.Tensor, tokens: torch.Tensor,
                       seqs: List[Sequence]):
        """Top-K logprobs (K = max over the batch; per-seq trim happens at
        commit). The sampled token's own logprob is appended as column K so
        it is always present even when outside the top-K."""
        K = max((s.sampling.logprobs or 0) for s in seqs)
        if K <= 0:
            return tokens, None, None
        lp = torch.log_softmax(logits.float(), dim=-1)
        top_vals, top_ids = torch.topk(lp, K, dim=-1)
        own = lp.gather(1, tokens.unsqueeze(1))
        top_vals = torch.cat([top_vals, own], dim=1)
        top_ids = torch.cat([top_ids, tokens.unsqueeze(1)], dim=1)
        return tokens, top_vals, top_ids
