import argparse
import dataclasses, os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from kaito_amd.engine import EngineConfig, LLMEngine, SamplingParams
from kaito_amd.models import get_model_config
from kaito_amd.parallel.state import init_parallel
init_parallel(1)
ap = argparse.ArgumentParser()
ap.add_argument("--model", default="llama-3-8b")
ap.add_argument("--tokens", type=int, default=256)
args = ap.parse_args()
for quant in ("", "w4a16"):
    mc = dataclasses.replace(get_model_config(args.model), quant_method=quant)
    cfg = EngineConfig(model=mc, max_num_seqs=8, max_model_len=512,
                       graph_batch_sizes=(1, 2, 4, 8))
    eng = LLMEngine(cfg)
    eng.capture_graphs()
    sp = SamplingParams(max_tokens=args.tokens, ignore_eos=True)
    eng.generate([list(range(100, 164))], sp)          # warmup
    torch.cuda.synchronize(); t0 = time.perf_counter()
    eng.generate([list(range(200, 264))], sp)
    eng.flush(); torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    label = quant or "bf16"
    n = args.tokens
    print(f"single-stream {args.model} {label:6s}: {n/dt:7.1f} tok/s  "
          f"({dt/n*1000:.2f} ms/token)")
    del eng
    torch.cuda.empty_cache()
