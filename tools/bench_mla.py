#!/usr/bin/env python3
"""MLA decode kernel microbench (+ occupancy sweep via KAITO_MLA_OCC)."""
import os
import subprocess
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

SHAPES = [(768, 300), (256, 400), (1024, 300)]   # (batch, kv_len)
H, R_, P_, BS = 16, 512, 64, 16


def run_one():
    import torch
    from kaito_amd import ops
    from kaito_amd.ops import torch_ref as TR

    torch.manual_seed(0)
    dev = "cuda"
    tag = os.environ.get("KAITO_MLA_OCC", "dflt") + \
        ("+db" if os.environ.get("KAITO_MLA_DB") else "") + \
        ("+ph" + os.environ.get("KAITO_MLA_PH")
         if os.environ.get("KAITO_MLA_PH") else "") + \
        ("+mf" if os.environ.get("KAITO_MLA_MF") else "") + \
        ("+mv" if os.environ.get("KAITO_MLA_MV") else "")
    DT = R_ + P_
    first = os.environ.get("KAITO_MLA_PH", "3") == "3"
    for bs, L in SHAPES:
        bps = (L + BS - 1) // BS
        nblocks = bs * bps + 1
        cache = (torch.randn(nblocks, BS, DT, device=dev) * 0.5).to(torch.bfloat16)
        q = (torch.randn(bs, H, DT, device=dev) * 0.3).to(torch.bfloat16)
        bt = torch.arange(1, bs * bps + 1, dtype=torch.int32,
                          device=dev).reshape(bs, bps)
        sl = torch.full((bs,), L, dtype=torch.int32, device=dev)
        scale = 192 ** -0.5
        if first:   # one numerics check per variant
            got = ops.mla_decode(q, cache, bt, sl, scale, R_)
            want = TR.mla_decode(q.float(), cache.float(), bt, sl, scale, R_)
            err = (got.float() - want).abs().max().item()
            assert err < 2e-2, f"numerics {err}"
            first = False
        for _ in range(10):
            ops.mla_decode(q, cache, bt, sl, scale, R_)
        torch.cuda.synchronize()
        s, e = torch.cuda.Event(True), torch.cuda.Event(True)
        iters = 50
        s.record()
        for _ in range(iters):
            ops.mla_decode(q, cache, bt, sl, scale, R_)
        e.record()
        torch.cuda.synchronize()
        us = s.elapsed_time(e) / iters * 1000
        rd = bs * L * DT * 2
        print(f"[occ={tag:>4}] bs={bs:4d} L={L:4d}: {us:7.1f} us  "
              f"{rd / us / 1e6:.2f} TB/s cache-read", flush=True)


def main():
    if "--sweep" not in sys.argv:
        run_one()
        return
    # default (MF+MV) vs regression references
    variants = [{}, {"KAITO_MLA_MV": "0"},
                {"KAITO_MLA_MF": "0", "KAITO_MLA_MV": "0"}]
    for v in variants:
        env = {k: val for k, val in os.environ.items()
               if not k.startswith("KAITO_MLA")}
        env.update(v)
        subprocess.run([sys.executable, __file__], env=env, check=True)


if __name__ == "__main__":
    main()
