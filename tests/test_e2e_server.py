"""Subprocess e2e: the real server entrypoint on a TCP port (CPU, tiny
model) — the closest no-cluster analog of the reference's e2e preset tests
(test/e2e/preset_vllm_test.go)."""
import os
import signal
import socket
import subprocess
import sys
import time

import httpx
import pytest


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture(scope="module")
def server():
    port = _free_port()
    env = dict(os.environ)
    proc = subprocess.Popen(
        [sys.executable, "-m", "kaito_amd.server.entrypoint",
         "--model", "tiny-llama-test", "--port", str(port),
         "--host", "127.0.0.1", "--max-num-seqs", "8",
         "--max-model-len", "128", "--enforce-eager"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    base = f"http://127.0.0.1:{port}"
    try:
        deadline = time.monotonic() + 60
        while time.monotonic() < deadline:
            if proc.poll() is not None:
                out = proc.stdout.read().decode()
                raise RuntimeError(f"server died:\n{out[-2000:]}")
            try:
                if httpx.get(f"{base}/health", timeout=2).status_code == 200:
                    break
            except httpx.HTTPError:
                time.sleep(0.5)
        else:
            raise TimeoutError("server never became healthy")
        yield base
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()


def test_e2e_completion_roundtrip(server):
    r = httpx.post(f"{server}/v1/completions", json={
        "prompt": [5, 6, 7, 8], "max_tokens": 4, "temperature": 0,
        "ignore_eos": True}, timeout=30)
    assert r.status_code == 200
    assert r.json()["usage"]["completion_tokens"] == 4


def test_e2e_metrics_surface(server):
    body = httpx.get(f"{server}/metrics", timeout=10).text
    assert "vllm:cache_config_info" in body
    assert "vllm:generation_tokens" in body


def test_e2e_benchmark_probe(server):
    """The self-benchmark entrypoint runs against the live server and emits
    the controller-ingestible result lines."""
    out = subprocess.run(
        [sys.executable, "-m", "kaito_amd.server.benchmark_entrypoint",
         "--base-url", server, "--duration", "3", "--input-tokens", "8",
         "--output-tokens", "4", "--max-concurrency", "2",
         "--health-timeout", "30"],
        capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr[-500:]
    assert "KAITO_BENCHMARK_RESULT" in out.stdout
    assert "peakTokensPerMinute" in out.stdout


def test_dp_frontend_two_replicas():
    """--data-parallel-size 2 spawns two engine replicas behind the
    front-end proxy: /health aggregates, completions round-robin, and
    /metrics sums counters across replicas."""
    port = _free_port()
    proc = subprocess.Popen(
        [sys.executable, "-m", "kaito_amd.server.entrypoint",
         "--model", "tiny-llama-test", "--port", str(port),
         "--host", "127.0.0.1", "--max-num-seqs", "8",
         "--max-model-len", "128", "--enforce-eager",
         "--data-parallel-size", "2"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    base = f"http://127.0.0.1:{port}"
    try:
        deadline = time.monotonic() + 120
        up = False
        while time.monotonic() < deadline:
            if proc.poll() is not None:
                out = proc.stdout.read().decode()
                raise RuntimeError(f"dp frontend died:\n{out[-2000:]}")
            try:
                if httpx.get(base + "/health", timeout=2).status_code == 200:
                    up = True
                    break
            except Exception:  # noqa: BLE001
                pass
            time.sleep(0.5)
        assert up, "front-end never became healthy"
        for _ in range(4):
            r = httpx.post(base + "/v1/completions", json={
                "prompt": [3, 14, 15], "max_tokens": 4,
                "temperature": 0.0, "ignore_eos": True}, timeout=60)
            assert r.status_code == 200
            assert r.json()["usage"]["completion_tokens"] == 4
        m = httpx.get(base + "/metrics", timeout=10).text
        # counters summed across BOTH replicas: 4 requests x 4 tokens
        for line in m.splitlines():
            if line.startswith("vllm:generation_tokens_total "):
                assert float(line.split()[-1]) == 16.0
                break
        else:
            raise AssertionError("missing generation_tokens_total")
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=20)
        except subprocess.TimeoutExpired:
            proc.kill()
