"""Data-parallel serving front-end.

The planner's tier 1 (model < 50% of one 288 GiB MI355X —
operator/planner.py:50) gives one pod ALL the node's GPUs with
data-parallel-size=N: N independent TP=1 engine replicas behind one
OpenAI endpoint (the reference gets this from vLLM's DP engine-core
replicas; SURVEY.md §2.3 "DP" row).

MI355X shape: one engine PROCESS per GPU (HIP_VISIBLE_DEVICES pinning —
no multi-GPU process state), a parent asyncio reverse proxy on :5000
that routes requests least-outstanding-first, aggregates /health (all
replicas) and sums /metrics counters across replicas so the benchmark
probe and KEDA see whole-pod numbers.
"""
from __future__ import annotations

import os
import re
import subprocess
import sys
import time
from typing import Dict, List, Optional

import httpx
from fastapi import FastAPI, Request, Response

PROXY_TIMEOUT_S = 600.0


def spawn_replicas(argv: List[str], dp: int, base_port: int,
                   python: Optional[str] = None) -> List[subprocess.Popen]:
    """Launch dp engine replicas of the entrypoint, one per GPU, on
    base_port+1+i. argv is the original CLI minus the dp/port flags."""
    procs = []
    for i in range(dp):
        env = dict(os.environ)
        env["HIP_VISIBLE_DEVICES"] = str(i)
        env["CUDA_VISIBLE_DEVICES"] = str(i)
        env["KAITO_DP_RANK"] = str(i)
        cmd = [python or sys.executable, "-m", "kaito_amd.server.entrypoint",
               *argv, "--data-parallel-size", "1",
               "--port", str(base_port + 1 + i)]
        procs.append(subprocess.Popen(cmd, env=env))
    return procs


def _merge_metrics(texts: List[str]) -> str:
    """Sum prometheus samples with identical name+labels across replicas
    (counters/gauges; HELP/TYPE lines kept once)."""
    meta: List[str] = []
    seen_meta = set()
    sums: Dict[str, float] = {}
    order: List[str] = []
    for t in texts:
        for line in t.splitlines():
            if line.startswith("#"):
                if line not in seen_meta:
                    seen_meta.add(line)
                    meta.append(line)
                continue
            m = re.match(r"^(.*\S)\s+([-+0-9.eEnaif]+)$", line)
            if not m:
                continue
            key, val = m.group(1), m.group(2)
            try:
                v = float(val)
            except ValueError:
                continue
            if key not in sums:
                sums[key] = 0.0
                order.append(key)
            sums[key] += v
    body = meta + [f"{k} {sums[k]}" for k in order]
    return "\n".join(body) + "\n"


def build_dp_app(ports: List[int], procs=None) -> FastAPI:
    app = FastAPI(title="kaito-amd DP front-end")
    outstanding = [0] * len(ports)
    client = httpx.AsyncClient(timeout=PROXY_TIMEOUT_S)
    app.state.outstanding = outstanding

    @app.get("/health")
    async def health():
        for p in ports:
            try:
                r = await client.get(f"http://127.0.0.1:{p}/health",
                                     timeout=5.0)
                if r.status_code != 200:
                    return Response(status_code=503)
            except Exception:  # noqa: BLE001
                return Response(status_code=503)
        return {"status": "ok", "replicas": len(ports)}

    @app.get("/metrics")
    async def metrics():
        texts = []
        for p in ports:
            try:
                r = await client.get(f"http://127.0.0.1:{p}/metrics",
                                     timeout=5.0)
                texts.append(r.text)
            except Exception:  # noqa: BLE001
                continue
        return Response(_merge_metrics(texts), media_type="text/plain")

    @app.api_route("/{path:path}",
                   methods=["GET", "POST", "PUT", "DELETE"])
    async def proxy(path: str, request: Request):
        body = await request.body()
        headers = {k: v for k, v in request.headers.items()
                   if k.lower() not in ("host", "content-length")}
        params = dict(request.query_params)
        last_exc = None
        tried = set()
        # a dead/unreachable replica must not take the endpoint down:
        # retry the request on each remaining replica once
        for _ in range(len(ports)):
            i = min((j for j in range(len(ports)) if j not in tried),
                    key=lambda j: outstanding[j])
            tried.add(i)
            outstanding[i] += 1
            try:
                r = await client.request(
                    request.method, f"http://127.0.0.1:{ports[i]}/{path}",
                    content=body, headers=headers, params=params)
                return Response(r.content, status_code=r.status_code,
                                media_type=r.headers.get("content-type"))
            except httpx.TransportError as e:
                last_exc = e
            finally:
                outstanding[i] -= 1
        return Response(f"all replicas unreachable: {last_exc}",
                        status_code=503)
    return app


def wait_replicas_ready(ports: List[int], timeout_s: float = 1800.0) -> bool:
    deadline = time.monotonic() + timeout_s
    pending = set(ports)
    with httpx.Client(timeout=2.0) as c:
        while pending and time.monotonic() < deadline:
            for p in list(pending):
                try:
                    if c.get(f"http://127.0.0.1:{p}/health").status_code == 200:
                        pending.discard(p)
                except Exception:  # noqa: BLE001
                    pass
            if pending:
                time.sleep(1.0)
    return not pending


def serve_dp(argv: List[str], dp: int, host: str, port: int) -> None:
    """Entry: spawn replicas, wait ready, run the proxy (blocking)."""
    import uvicorn
    procs = spawn_replicas(argv, dp, port)
    ports = [port + 1 + i for i in range(dp)]
    try:
        if not wait_replicas_ready(ports):
            raise SystemExit("DP replicas failed to become ready")
        uvicorn.run(build_dp_app(ports, procs), host=host, port=port,
                    log_level="warning")
    finally:
        for pr in procs:
            pr.terminate()
        for pr in procs:
            try:
                pr.wait(timeout=10)
            except subprocess.TimeoutExpired:
                pr.kill()
