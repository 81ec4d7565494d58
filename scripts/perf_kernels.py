#!/usr/bin/env python3
"""Kernel microbenchmarks on MI355X: prints achieved bandwidth/time for the
hot kernels at bench-like shapes. Run under gpurun."""
import argparse
import math
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import kaito_amd.ops as ops


def bench(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--bs", type=int, default=512)
    p.add_argument("--seqlen", type=int, default=400)
    args = p.parse_args()
    ops.load_extension()
    dev = "cuda"
    T, KH, G, D, BS = args.bs, 8, 4, 128, 16
    QH = KH * G
    L = args.seqlen
    mb = (L + BS - 1) // BS
    NB = T * mb + 1
    kc = torch.randn(NB, KH, BS, D, device=dev).to(torch.bfloat16)
    vc = torch.randn_like(kc)
    bt = torch.arange(1, T * mb + 1, dtype=torch.int32,
                      device=dev).reshape(T, mb)
    q = torch.randn(T, QH, D, device=dev).to(torch.bfloat16)
    sl = torch.full((T,), L, dtype=torch.int32, device=dev)
    scale = 1.0 / math.sqrt(D)

    out = torch.empty_like(q)
    t = bench(lambda: torch.ops.kaito.paged_attention(
        out, q, kc, vc, bt, sl, scale))
    bytes_moved = T * L * KH * D * 2 * 2  # K+V read
    print(f"paged_attention    bs={T} L={L}: {t*1e6:.1f} us  "
          f"{bytes_moved/t/1e12:.2f} TB/s")
    empty_sinks = torch.empty(0, dtype=torch.float32, device=dev)
    t = bench(lambda: torch.ops.kaito.paged_attention_sp(
        out, q, kc, vc, bt, sl, scale, 0, empty_sinks))
    print(f"paged_attention_sp bs={T} L={L}: {t*1e6:.1f} us  "
          f"{bytes_moved/t/1e12:.2f} TB/s")

    # diagnostics: mode0 = pure read; mode1 = +dot+accumulate (no shfl)
    outd = torch.empty(T * KH, dtype=torch.float32, device=dev)
    t = bench(lambda: torch.ops.kaito.paged_read_bw(outd, kc, vc, bt, sl, 0))
    print(f"paged_read_bw  bs={T} L={L}: {t*1e6:.1f} us  "
          f"{bytes_moved/t/1e12:.2f} TB/s")
    t = bench(lambda: torch.ops.kaito.paged_read_bw(outd, kc, vc, bt, sl, 1))
    print(f"paged_dot_bw   bs={T} L={L}: {t*1e6:.1f} us  "
          f"{bytes_moved/t/1e12:.2f} TB/s")

    # rmsnorm
    x = torch.randn(T, 4096, device=dev).to(torch.bfloat16)
    r = torch.randn_like(x)
    w = torch.ones(4096, device=dev).to(torch.bfloat16)
    t = bench(lambda: ops.fused_add_rms_norm(x, r, w, 1e-5))
    print(f"fused_add_rms_norm [{T},4096]: {t*1e6:.1f} us  "
          f"{(T*4096*2*4)/t/1e12:.2f} TB/s")

    # silu
    g = torch.randn(T, 2 * 14336, device=dev).to(torch.bfloat16)
    t = bench(lambda: ops.silu_and_mul(g))
    print(f"silu_and_mul [{T},2x14336]: {t*1e6:.1f} us  "
          f"{(T*14336*2*3)/t/1e12:.2f} TB/s")

    # prefill attention 8192 tokens
    Tp = 8192
    qp = torch.randn(Tp, QH, D, device=dev).to(torch.bfloat16)
    kp = torch.randn(Tp, KH, D, device=dev).to(torch.bfloat16)
    vp = torch.randn_like(kp)
    cu = torch.arange(0, Tp + 1, 256, dtype=torch.int32, device=dev)
    t = bench(lambda: ops.prefill_attention(qp, kp, vp, cu, scale), iters=20)
    flops = 0
    n = 256
    flops = (Tp // n) * (2 * 2 * QH * n * n * D) / 2  # causal half
    print(f"prefill_attn {Tp} tok (seqs of 256): {t*1e3:.2f} ms  "
          f"{flops/t/1e12:.0f} TFLOP/s")


if __name__ == "__main__":
    main()
