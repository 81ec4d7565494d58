#!/usr/bin/env python3
"""MLA kernel debugging harness: per-case error stats vs the fp32 ref."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402
from kaito_amd import ops  # noqa: E402
from kaito_amd.ops import torch_ref as R  # noqa: E402

DEV = "cuda"
R_, P_, BS = 512, 64, 16
DT = R_ + P_


def run_case(T, H, lens, tag):
    torch.manual_seed(7)
    nb = [(x + BS - 1) // BS for x in lens]
    NB = sum(nb) + 1
    cache = (torch.randn(NB, BS, DT, device=DEV)).to(torch.bfloat16)
    bt = torch.zeros(T, max(nb), dtype=torch.int32, device=DEV)
    nxt = 1
    for i in range(T):
        bt[i, :nb[i]] = torch.arange(nxt, nxt + nb[i], dtype=torch.int32,
                                     device=DEV)
        nxt += nb[i]
    sl = torch.tensor(lens, dtype=torch.int32, device=DEV)
    q = (torch.randn(T, H, DT, device=DEV) * 0.3).to(torch.bfloat16)
    scale = 192 ** -0.5
    out = ops.mla_decode(q, cache, bt, sl, scale, R_)
    ref = R.mla_decode(q.cpu().float(), cache.cpu().float(), bt.cpu(),
                       sl.cpu(), scale, R_).to(DEV)
    err = (out.float() - ref).abs()
    rel = err / (ref.abs() + 1e-3)
    per_seq = err.amax(dim=(1, 2))
    per_head = err.amax(dim=(0, 2))
    print(f"[{tag}] max={err.max().item():.4f} mean={err.mean().item():.5f} "
          f"bad frac={(err > 2e-2).float().mean().item():.4f}")
    print(f"   per-seq max: {[round(x, 3) for x in per_seq.tolist()]}")
    print(f"   per-head max: {[round(x, 3) for x in per_head.tolist()][:16]}")
    # dim-slice pattern: is the error localized to certain 32-dim slices?
    per_slice = err.reshape(T, -1, 16, 32).amax(dim=(0, 1, 3))
    print(f"   per-dimslice max: {[round(x, 3) for x in per_slice.tolist()]}")


run_case(1, 16, [16], "1seq-1block")
run_case(1, 16, [20], "1seq-2block")
run_case(1, 16, [8], "1seq-partial")
run_case(5, 16, [200, 33, 7, 390, 64], "case1")
run_case(3, 4, [100, 17, 255], "case2-padheads")
run_case(2, 32, [48, 312], "case3-2tiles")
