"""Inference preset entrypoint — the MI355X analog of the reference's
presets/workspace/inference/vllm/inference_api.py:

  * KAITO argparse layered over engine args (reference :64-159)
  * --kaito-config-file YAML merge (reference :128-154)
  * free-VRAM probe → gpu_memory_utilization / max-model-len auto
    (reference :439-496 does this in a throwaway subprocess; here the probe
    is hipMemGetInfo inside ModelRunner.profile_and_allocate_kv)
  * serves OpenAI API + /metrics on :5000

Launch:  python -m kaito_amd.server.entrypoint --model llama-3-8b \
            --tensor-parallel-size 1 [--kaito-config-file /mnt/config/inference_config.yaml]
"""
from __future__ import annotations

import argparse
import logging
import os
import sys

import yaml

logger = logging.getLogger("kaito_amd.server")


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="kaito-amd inference engine")
    p.add_argument("--model", default=os.environ.get("KAITO_MODEL", "llama-3-8b"))
    p.add_argument("--weights-path", default=os.environ.get("KAITO_WEIGHTS_PATH"))
    p.add_argument("--kaito-config-file", default=None,
                   help="YAML config merged under vllm:/engine: key")
    p.add_argument("--tensor-parallel-size", type=int, default=1)
    p.add_argument("--data-parallel-size", type=int, default=1)
    p.add_argument("--pipeline-parallel-size", type=int, default=1)
    p.add_argument("--max-model-len", default="auto")
    p.add_argument("--max-num-seqs", type=int, default=256)
    p.add_argument("--gpu-memory-utilization", type=float, default=0.90)
    p.add_argument("--swap-space", type=float, default=0.0,
                   help="GiB of host RAM for KV offload (LMCache analog)")
    p.add_argument("--kv-cache-dtype", default="auto")
    p.add_argument("--enable-kv-events", action="store_true",
                   help="publish BlockStored/BlockRemoved on :5557 "
                        "(EPP KVCache-aware routing)")
    p.add_argument("--kv-events-port", type=int, default=5557)
    p.add_argument("--enable-lora", action="store_true")
    p.add_argument("--quantization", default="",
                   choices=["", "w4a16", "awq"],
                   help="weight-only 4-bit serving (HIP GEMV/dequant+MFMA)")
    p.add_argument("--enforce-eager", action="store_true")
    p.add_argument("--port", type=int, default=int(os.environ.get("PORT", 5000)))
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--served-model-name", default=None)
    p.add_argument("--tokenizer", default=None)
    return p


def merge_config_file(args: argparse.Namespace, path: str | None) -> None:
    """Merge the Workspace inference ConfigMap (reference semantics: file
    values override defaults, CLI flags override the file)."""
    if not path or not os.path.exists(path):
        return
    explicit = {a.lstrip("-").replace("-", "_").split("=")[0]
                for a in sys.argv[1:] if a.startswith("--")}
    with open(path) as f:
        data = yaml.safe_load(f) or {}
    section = data.get("vllm") or data.get("engine") or {}
    for k, v in section.items():
        attr = k.replace("-", "_")
        if hasattr(args, attr) and attr not in explicit:
            setattr(args, attr, v)


def main(argv=None):
    logging.basicConfig(level=logging.INFO)
    args = build_parser().parse_args(argv)
    merge_config_file(args, args.kaito_config_file)

    if args.data_parallel_size > 1:
        # tier-1 DP pod: one engine process per GPU behind a front-end
        # proxy (planner.py:50 gives sub-144GiB models all 8 GPUs as
        # independent TP=1 replicas)
        from .dp_frontend import serve_dp
        raw = list(argv if argv is not None else sys.argv[1:])
        child = []
        skip = False
        for i, a in enumerate(raw):
            if skip:
                skip = False
                continue
            if a in ("--data-parallel-size", "--port"):
                skip = True
                continue
            if a.startswith("--data-parallel-size=") or \
                    a.startswith("--port="):
                continue
            child.append(a)
        return serve_dp(child, args.data_parallel_size, args.host,
                        args.port)

    import torch
    from ..engine import EngineConfig, LLMEngine
    from ..models import get_model_config
    from ..parallel.state import init_parallel
    from .async_engine import AsyncLLMEngine
    from .api import build_app
    from .tokenizer import load_tokenizer
    from . import metrics

    mc = get_model_config(args.model, args.weights_path)
    if args.quantization:
        import dataclasses
        mc = dataclasses.replace(mc, quant_method=args.quantization)
    if mc.runtime == "transformers":
        # non-llama-family architecture (falcon/gemma-3/deepseek-MLA/
        # gpt-oss): serve via the fallback runtime — the reference's
        # vLLM vs text-generation runtime split (supported_models.yaml
        # `runtime: tfs` rows).
        from .transformers_runtime import main as tfs_main
        return tfs_main(["--model", args.model, "--port", str(args.port),
                         "--host", args.host] +
                        (["--weights-path", args.weights_path]
                         if args.weights_path else []))
    init_parallel(tp_size=args.tensor_parallel_size,
                  pp_size=args.pipeline_parallel_size)
    max_len = None if str(args.max_model_len) == "auto" else int(args.max_model_len)
    cfg = EngineConfig(
        model=mc,
        device="cuda" if torch.cuda.is_available() else "cpu",
        max_num_seqs=args.max_num_seqs,
        max_model_len=max_len,
        gpu_memory_utilization=args.gpu_memory_utilization,
        tensor_parallel_size=args.tensor_parallel_size,
        enforce_eager=args.enforce_eager,
        enable_lora=args.enable_lora,
        kv_offload=args.swap_space > 0,
        kv_offload_bytes=int(args.swap_space * (1 << 30)) or None,
        kv_cache_dtype=("fp8" if args.kv_cache_dtype in
                        ("fp8", "fp8_e4m3") else "auto"),
    )
    if args.kv_cache_dtype not in ("auto", "bf16", "bfloat16", "fp8",
                                   "fp8_e4m3"):
        raise SystemExit(f"kv-cache-dtype {args.kv_cache_dtype!r} not "
                         "supported (auto/bf16/fp8)")
    metrics.MODEL_DOWNLOAD_PROGRESS.set(0.0)
    engine = LLMEngine(cfg, weights_path=args.weights_path)
    if args.enable_kv_events:
        from ..engine.kv_events import KVEventPublisher
        engine.kv_publisher = KVEventPublisher(port=args.kv_events_port)
    metrics.MODEL_DOWNLOAD_PROGRESS.set(1.0)
    metrics.MODEL_DOWNLOAD_DONE.set(1)
    if not args.enforce_eager and torch.cuda.is_available():
        engine.capture_graphs()
    tokenizer = load_tokenizer(args.tokenizer or args.weights_path,
                               mc.vocab_size)
    eos = getattr(tokenizer, "eos_token_id", None)
    engine.eos_token_id = eos
    async_engine = AsyncLLMEngine(engine).start()
    app = build_app(async_engine, tokenizer,
                    args.served_model_name or args.model)

    import uvicorn
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
