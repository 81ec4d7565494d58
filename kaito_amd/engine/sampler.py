"""Token sampling: greedy / temperature / top-k / top-p on-device.

Operates on the logits of the batch's last tokens. Torch-native (runs on
GPU); a dedicated HIP sampling kernel is a later optimization (the sampler
is ~0 cost next to the decode forward).
"""
from __future__ import annotations

from typing import List

import torch

from .sequence import Sequence


class Sampler:
    def __init__(self, device: str, seed: int = 0):
        self.device = device
        self.gen = None
        if device != "cpu":
            self.gen = torch.Generator(device=device)
            self.gen.manual_seed(seed)

    @torch.no_grad()
    def sample(self, logits: torch.Tensor, seqs: List[Sequence]) -> torch.Tensor:
        """logits: [n, vocab] (float). Returns [n] long on logits.device.

        All branch decisions come from HOST-side sampling params — no
        device reads, so the call enqueues async (pipelined decode relies
        on sample() never synchronizing)."""
        temps_l = [s.sampling.temperature for s in seqs]
        if all(t <= 0.0 for t in temps_l):
            return logits.argmax(dim=-1)
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
        return torch.where(greedy, logits.argmax(dim=-1), sampled)
