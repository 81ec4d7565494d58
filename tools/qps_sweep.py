#!/usr/bin/env python3
"""QPS-sweep serving benchmark — the reference's published methodology
(website/docs/gpu-benchmarks.md: guidellm, Poisson request arrivals,
60 s per rate, input 200 / output 200 tokens) against the native engine.

Emits one CSV row per QPS with the columns of the reference's summary
tables (benchmark-phi4-p1.csv: TTFT and ITL mean/median/p99) plus output
tok/s, so results are DIRECTLY comparable to BASELINE.md's A100/A10 rows.

Usage (on a GPU box):
  python tools/qps_sweep.py --model phi-4-mini --qps 1,2,4,8,16,32,64 \
      --duration 60 --out profiles/qps_sweep.csv
"""
from __future__ import annotations

import argparse
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch


def pct(vals, p):
    if not vals:
        return float("nan")
    return float(np.percentile(np.array(vals), p))


def run_rate(eng, SamplingParams, qps: float, duration: float,
             in_tok: int, out_tok: int, vocab: int, seed: int = 42):
    """Poisson open-loop load at `qps` for `duration` seconds; returns
    per-request metrics for requests that COMPLETED in the window."""
    rng = np.random.default_rng(seed)
    # pre-draw arrival times (exponential inter-arrivals)
    arrivals = []
    t = 0.0
    while t < duration:
        arrivals.append(t)
        t += rng.exponential(1.0 / qps)
    prompts = [rng.integers(10, vocab - 10, in_tok).tolist()
               for _ in range(len(arrivals))]
    sp = SamplingParams(max_tokens=out_tok, ignore_eos=True)

    sched = {}          # seq_id -> scheduled arrival (monotonic)
    finished = []       # (arrival, first_token_time, finish_time, n_out)
    next_i = 0
    t0 = time.monotonic()
    while True:
        now = time.monotonic() - t0
        while next_i < len(arrivals) and arrivals[next_i] <= now:
            sid = eng.add_request(prompts[next_i], sp)
            seq = eng.seqs[sid]
            seq.arrival_time = t0 + arrivals[next_i]  # true scheduled time
            sched[sid] = arrivals[next_i]
            next_i += 1
        done = eng.step()
        for s in done:
            if s.seq_id in sched:
                finished.append(s)
        if now >= duration and next_i >= len(arrivals):
            # drain what's in flight (counted toward latency percentiles)
            guard = time.monotonic() + 120
            while eng.has_unfinished() and time.monotonic() < guard:
                for s in eng.step():
                    if s.seq_id in sched:
                        finished.append(s)
            eng.flush()
            break
        if not eng.has_unfinished() and next_i < len(arrivals):
            time.sleep(max(0.0, arrivals[next_i] - (time.monotonic() - t0)))
    wall = time.monotonic() - t0

    ttft = [(s.first_token_time - s.arrival_time) * 1e3 for s in finished
            if s.first_token_time is not None]
    itl = []
    out_tokens = 0
    for s in finished:
        n = len(s.output_token_ids)
        out_tokens += n
        if n > 1 and s.finish_time and s.first_token_time:
            itl.append((s.finish_time - s.first_token_time) * 1e3 / (n - 1))
    return {
        "qps": qps,
        "requests": len(finished),
        "duration_s": round(wall, 1),
        "output_tok_s_mean": round(out_tokens / wall, 1),
        "ttft_mean_ms": round(statistics.fmean(ttft), 1) if ttft else None,
        "ttft_p50_ms": round(pct(ttft, 50), 1),
        "ttft_p99_ms": round(pct(ttft, 99), 1),
        "itl_mean_ms": round(statistics.fmean(itl), 2) if itl else None,
        "itl_p50_ms": round(pct(itl, 50), 2),
        "itl_p99_ms": round(pct(itl, 99), 2),
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="phi-4-mini-instruct")
    p.add_argument("--qps", default="1,2,4,8,16,32,64")
    p.add_argument("--duration", type=float, default=60.0)
    p.add_argument("--in-tokens", type=int, default=200)
    p.add_argument("--out-tokens", type=int, default=200)
    p.add_argument("--max-num-seqs", type=int, default=1024)
    p.add_argument("--eager", action="store_true")
    p.add_argument("--out", default="")
    args = p.parse_args()

    from kaito_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from kaito_amd.models import get_model_config
    from kaito_amd.parallel.state import init_parallel
    init_parallel(1)
    mc = get_model_config(args.model)
    cfg = EngineConfig(
        model=mc, device="cuda" if torch.cuda.is_available() else "cpu",
        max_num_seqs=args.max_num_seqs,
        max_model_len=args.in_tokens + args.out_tokens + 16,
        enforce_eager=args.eager)
    eng = LLMEngine(cfg)
    if not args.eager:
        eng.capture_graphs()

    cols = ["qps", "requests", "duration_s", "output_tok_s_mean",
            "ttft_mean_ms", "ttft_p50_ms", "ttft_p99_ms",
            "itl_mean_ms", "itl_p50_ms", "itl_p99_ms"]
    lines = ["# model=%s in=%d out=%d poisson %.0fs/rate (reference "
             "methodology: website/docs/gpu-benchmarks.md)"
             % (args.model, args.in_tokens, args.out_tokens, args.duration),
             ",".join(cols)]
    print(lines[0])
    print(lines[1], flush=True)
    for q in [float(x) for x in args.qps.split(",")]:
        r = run_rate(eng, SamplingParams, q, args.duration,
                     args.in_tokens, args.out_tokens, mc.vocab_size)
        row = ",".join(str(r[c]) for c in cols)
        lines.append(row)
        print(row, flush=True)
    if args.out:
        with open(args.out, "w") as f:
            f.write("\n".join(lines) + "\n")


if __name__ == "__main__":
    main()
