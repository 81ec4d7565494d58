"""KV-cache host offload — the MI355X-native take on the LMCache CPU
offload the reference wires up (inference_api.py:514-544: chunk size 256,
RAM budget = avail * util / TP, connector kv_both).

Mechanics:
  * finished sequences' KV blocks are copied GPU→pinned-host on a side HIP
    stream (hipMemcpyAsync under torch .copy_(non_blocking=True)),
    chunked at CHUNK_TOKENS (=256) and keyed by the sha of the covered
    token prefix (chain-hash like LMCache);
  * a resubmitted prompt whose FULL token sequence matches a cached chain
    restores host→GPU into freshly allocated blocks and skips prefill
    entirely — the engine then runs one decode step from the last prompt
    token (its KV slot is rewritten with identical values);
  * a prompt sharing only a PREFIX with a cached chain restores the
    longest common prefix (restore_prefix) and the scheduler prefills
    just the suffix as a chunked tail over the restored KV (the paged
    context-prefill path, ops/csrc/context_attention.hip);
  * LRU eviction under a byte budget (default: 0.5 of available RAM,
    the reference's LMCACHE default utilization).
"""
from __future__ import annotations

import hashlib
from collections import OrderedDict
from typing import List, Optional, Tuple

import torch

CHUNK_TOKENS = 256


def _chain_hash(tokens: List[int]) -> str:
    return hashlib.sha256(
        b"|".join(str(t).encode() for t in tokens)).hexdigest()


class KVOffloadManager:
    def __init__(self, kv_caches, block_size: int,
                 max_bytes: Optional[int] = None, device: str = "cpu"):
        self.kv_caches = kv_caches            # [(k,v)] per layer, GPU
        self.block_size = block_size
        self.is_gpu = device != "cpu" and torch.cuda.is_available()
        if max_bytes is None:
            try:
                import psutil
                max_bytes = int(psutil.virtual_memory().available * 0.5)
            except Exception:
                max_bytes = 8 << 30
        self.max_bytes = max_bytes
        self.used_bytes = 0
        # key → (token_tuple, [per-layer (k,v) host tensors]); lookups match
        # any stored entry whose token sequence has the prompt as a prefix
        self._store: "OrderedDict[str, Tuple[tuple, list]]" = OrderedDict()
        self._stream = torch.cuda.Stream() if self.is_gpu else None
        self.hits = 0
        self.misses = 0

    # ------------------------------------------------------------- helpers
    def _seq_bytes(self, num_tokens: int) -> int:
        k0 = self.kv_caches[0][0]
        kvh, _, d = k0.shape[1], k0.shape[2], k0.shape[3]
        return (len(self.kv_caches) * 2 * num_tokens * kvh * d
                * k0.element_size())

    def _gather_tokens(self, block_table: List[int], num_tokens: int):
        """Copy the first num_tokens of a sequence's KV to host tensors
        [layers][2][kvh, num_tokens, d]."""
        bs = self.block_size
        nb = (num_tokens + bs - 1) // bs
        blocks = torch.tensor(block_table[:nb], dtype=torch.long,
                              device=self.kv_caches[0][0].device)
        out = []
        ctx = torch.cuda.stream(self._stream) if self._stream else _nullctx()
        with ctx:
            for (kc, vc) in self.kv_caches:
                # [nb, kvh, bs, d] → [kvh, nb*bs, d] → [:, :num_tokens]
                kb = kc[blocks].permute(1, 0, 2, 3).reshape(
                    kc.shape[1], nb * bs, kc.shape[3])[:, :num_tokens]
                vb = vc[blocks].permute(1, 0, 2, 3).reshape(
                    vc.shape[1], nb * bs, vc.shape[3])[:, :num_tokens]
                kh = torch.empty_like(kb, device="cpu",
                                      pin_memory=self.is_gpu)
                vh = torch.empty_like(vb, device="cpu",
                                      pin_memory=self.is_gpu)
                kh.copy_(kb, non_blocking=self.is_gpu)
                vh.copy_(vb, non_blocking=self.is_gpu)
                out.append((kh, vh))
        if self._stream:
            self._stream.synchronize()
        return out

    # ------------------------------------------------------------- offload
    def offload(self, tokens: List[int], block_table: List[int]) -> bool:
        """Store the KV of `tokens` (prompt+generated) under its chain key.
        Called on sequence finish, BEFORE blocks are freed."""
        n = len(tokens)
        if n < 1 or not block_table:
            return False
        need = self._seq_bytes(n)
        if need > self.max_bytes:
            return False
        key = _chain_hash(tokens)
        if key in self._store:
            self._store.move_to_end(key)
            return True
        while self.used_bytes + need > self.max_bytes and self._store:
            _, (old_t, _t) = self._store.popitem(last=False)  # LRU
            self.used_bytes -= self._seq_bytes(len(old_t))
        host = self._gather_tokens(block_table, n)
        self._store[key] = (tuple(tokens), host)
        self.used_bytes += need
        return True

    # ------------------------------------------------------------- restore
    def lookup(self, prompt: List[int]) -> Optional[str]:
        """Find a stored entry whose token sequence has `prompt` as a
        prefix (covers the whole prompt)."""
        p = tuple(prompt)
        L = len(p)
        for key, (toks, _) in self._store.items():
            if len(toks) >= L and toks[:L] == p:
                return key
        return None

    MIN_PREFIX = 16  # don't bother restoring less than one block

    def longest_prefix(self, prompt: List[int]) -> Tuple[Optional[str], int]:
        """Entry with the longest exact common prefix with `prompt`;
        returns (key, covered_tokens)."""
        best_key, best_n = None, 0
        p = tuple(prompt)
        for key, (toks, _) in self._store.items():
            n = 0
            for a, b in zip(p, toks):
                if a != b:
                    break
                n += 1
            if n > best_n:
                best_key, best_n = key, n
        return best_key, best_n

    def restore_prefix(self, prompt: List[int],
                       block_table: List[int]) -> int:
        """Restore the longest matching prefix into the sequence's blocks.
        Returns the covered token count (0 = miss / below threshold).
        Partially-restored boundary blocks are safe: the suffix prefill
        rewrites its own slots in stream order before attention reads."""
        key, covered = self.longest_prefix(prompt)
        if key is None or covered < self.MIN_PREFIX:
            self.misses += 1
            return 0
        self._store.move_to_end(key)
        self._restore_tokens(key, covered, block_table)
        self.hits += 1
        return covered

    def _restore_tokens(self, key: str, n: int,
                        block_table: List[int]) -> None:
        bs = self.block_size
        nb = (n + bs - 1) // bs
        dev = self.kv_caches[0][0].device
        blocks = torch.tensor(block_table[:nb], dtype=torch.long, device=dev)
        _toks, host = self._store[key]
        ctx = torch.cuda.stream(self._stream) if self._stream else _nullctx()
        with ctx:
            for (kc, vc), (kh, vh) in zip(self.kv_caches, host):
                kvh, d = kc.shape[1], kc.shape[3]
                pad = nb * bs
                kg = torch.zeros(kvh, pad, d, dtype=kc.dtype, device="cpu")
                vg = torch.zeros_like(kg)
                kg[:, :n] = kh[:, :n]
                vg[:, :n] = vh[:, :n]
                kdev = kg.to(dev, non_blocking=self.is_gpu)
                vdev = vg.to(dev, non_blocking=self.is_gpu)
                kc[blocks] = kdev.reshape(kvh, nb, bs, d).permute(1, 0, 2, 3)
                vc[blocks] = vdev.reshape(kvh, nb, bs, d).permute(1, 0, 2, 3)
        if self._stream:
            self._stream.synchronize()

    def restore(self, prompt: List[int], block_table: List[int]) -> bool:
        """Copy cached KV for `prompt` into the sequence's allocated blocks.
        Returns True on a full-prompt hit."""
        key = self.lookup(prompt)
        if key is None:
            self.misses += 1
            return False
        self._store.move_to_end(key)
        n = len(prompt)
        bs = self.block_size
        nb = (n + bs - 1) // bs
        dev = self.kv_caches[0][0].device
        blocks = torch.tensor(block_table[:nb], dtype=torch.long, device=dev)
        _toks, host = self._store[key]
        ctx = torch.cuda.stream(self._stream) if self._stream else _nullctx()
        with ctx:
            for (kc, vc), (kh, vh) in zip(self.kv_caches, host):
                kvh, d = kc.shape[1], kc.shape[3]
                pad = nb * bs
                kg = torch.zeros(kvh, pad, d, dtype=kc.dtype, device="cpu",
                                 pin_memory=False)
                vg = torch.zeros_like(kg)
                kg[:, :n] = kh[:, :n]
                vg[:, :n] = vh[:, :n]
                kdev = kg.to(dev, non_blocking=self.is_gpu)
                vdev = vg.to(dev, non_blocking=self.is_gpu)
                # [kvh, nb*bs, d] → [nb, kvh, bs, d]
                kc[blocks] = kdev.reshape(kvh, nb, bs, d).permute(1, 0, 2, 3)
                vc[blocks] = vdev.reshape(kvh, nb, bs, d).permute(1, 0, 2, 3)
        if self._stream:
            self._stream.synchronize()
        self.hits += 1
        return True


class _nullctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False
