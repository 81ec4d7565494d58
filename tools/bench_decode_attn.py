#!/usr/bin/env python3
"""Decode paged-attention microbench (the VERDICT #2 ladder).

Measures achieved K+V read bandwidth of the split-phase kernel at the
three canonical shapes (profiles/r02_perf_notes.md), optionally across
occupancy variants (KAITO_PA_SP_OCC=3|4|5 — the env is read once per
process by the extension, so each variant runs in a subprocess).

Usage:  python tools/bench_decode_attn.py            # current env variant
        python tools/bench_decode_attn.py --sweep-occ
"""
import argparse
import os
import subprocess
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

SHAPES = [(512, 400), (1024, 300), (256, 1500)]  # (batch, kv_len)
KH, G, D, BS = 8, 4, 128, 16


def run_one():
    import torch
    from kaito_amd import ops

    torch.manual_seed(0)
    dev = "cuda"
    occ = ",".join(f"{k[12:]}={v}" for k, v in os.environ.items()
                   if k.startswith("KAITO_PA_SP")) or "default"
    for bs, L in SHAPES:
        blocks_per_seq = (L + BS - 1) // BS
        nblocks = bs * blocks_per_seq + 1
        k_cache = torch.randn(nblocks, KH, BS, D, dtype=torch.bfloat16,
                              device=dev)
        v_cache = torch.randn_like(k_cache)
        q = torch.randn(bs, KH * G, D, dtype=torch.bfloat16, device=dev)
        bt = torch.arange(1, bs * blocks_per_seq + 1, dtype=torch.int32,
                          device=dev).reshape(bs, blocks_per_seq)
        sl = torch.full((bs,), L, dtype=torch.int32, device=dev)
        scale = D ** -0.5

        if bs == SHAPES[0][0]:  # one numerics check per variant
            from kaito_amd.ops import torch_ref
            got = ops.paged_attention(q, k_cache, v_cache, bt, sl, scale)
            want = torch_ref.paged_attention(
                q.float(), k_cache.float(), v_cache.float(), bt, sl, scale)
            err = (got.float() - want).abs().max().item()
            assert err < 2e-2, f"numerics mismatch: {err}"

        for _ in range(20):
            ops.paged_attention(q, k_cache, v_cache, bt, sl, scale)
        torch.cuda.synchronize()
        s, e = torch.cuda.Event(True), torch.cuda.Event(True)
        iters = 100
        s.record()
        for _ in range(iters):
            ops.paged_attention(q, k_cache, v_cache, bt, sl, scale)
        e.record()
        torch.cuda.synchronize()
        us = s.elapsed_time(e) / iters * 1000
        bytes_rd = bs * L * KH * D * 2 * 2  # K+V bf16
        print(f"[{occ:>7}] bs={bs:4d} L={L:4d}: {us:7.1f} us  "
              f"{bytes_rd / us / 1e6:.2f} TB/s", flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sweep-occ", action="store_true")
    args = ap.parse_args()
    if not args.sweep_occ:
        run_one()
        return
    variants = [{"KAITO_PA_SP_OCC": "4"},
                {"KAITO_PA_SP_UB": "2"},
                {"KAITO_PA_SP_OCC": "3"}]
    for v in variants:
        env = {k: val for k, val in os.environ.items()
               if not k.startswith("KAITO_PA_SP")}
        env.update(v)
        subprocess.run([sys.executable, __file__], env=env, check=True)


if __name__ == "__main__":
    main()
