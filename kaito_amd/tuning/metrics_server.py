"""Tuning metrics sidecar — parity with the reference's
presets/workspace/tuning/text-generation/metrics/metrics_server.py (GPU/CPU
utilisation + memory exposed for scraping during fine-tuning jobs; :112
uses gputil — ours reads torch.cuda + /proc)."""
from __future__ import annotations

import os

from fastapi import FastAPI


def collect_metrics() -> dict:
    out = {"cpu_percent": None, "mem_used_gib": None, "gpus": []}
    try:
        import psutil
        out["cpu_percent"] = psutil.cpu_percent(interval=0.0)
        out["mem_used_gib"] = round(
            psutil.virtual_memory().used / (1 << 30), 2)
    except ImportError:
        pass
    try:
        import torch
        if torch.cuda.is_available():
            for i in range(torch.cuda.device_count()):
                free, total = torch.cuda.mem_get_info(i)
                out["gpus"].append({
                    "index": i,
                    "mem_used_gib": round((total - free) / (1 << 30), 2),
                    "mem_total_gib": round(total / (1 << 30), 2),
                })
    except Exception:  # noqa: BLE001
        pass
    return out


def build_metrics_app() -> FastAPI:
    app = FastAPI(title="kaito-amd tuning metrics")

    @app.get("/metrics")
    async def metrics():
        m = collect_metrics()
        lines = []
        if m["cpu_percent"] is not None:
            lines.append(f"tuning_cpu_percent {m['cpu_percent']}")
        if m["mem_used_gib"] is not None:
            lines.append(f"tuning_mem_used_gib {m['mem_used_gib']}")
        for g in m["gpus"]:
            lines.append(f'tuning_gpu_mem_used_gib{{gpu="{g["index"]}"}} '
                         f'{g["mem_used_gib"]}')
        return "\n".join(lines) + "\n"

    @app.get("/health")
    async def health():
        return {"status": "ok"}
    return app


if __name__ == "__main__":
    import uvicorn
    uvicorn.run(build_metrics_app(), host="0.0.0.0",
                port=int(os.environ.get("METRICS_PORT", "8090")))


def main(argv=None):
    """Serve the tuning metrics sidecar (reference: metrics_server.py
    uvicorn on :5000)."""
    import argparse

    import uvicorn
    p = argparse.ArgumentParser()
    p.add_argument("--port", type=int, default=5000)
    p.add_argument("--host", default="0.0.0.0")
    args = p.parse_args(argv)
    uvicorn.run(build_metrics_app(), host=args.host, port=args.port,
                log_level="warning")
