"""Server tests: OpenAI API surface, SSE streaming, metrics names, 429
rate limiting — the contract the reference ecosystem (benchmark probe,
EPP, KEDA) scrapes.
"""
import threading
import time

import pytest
from fastapi.testclient import TestClient

from kaito_amd.engine import EngineConfig, LLMEngine
from kaito_amd.models import get_model_config
from kaito_amd.parallel.state import init_parallel
from kaito_amd.server.api import build_app
from kaito_amd.server.async_engine import AsyncLLMEngine
from kaito_amd.server.tokenizer import ByteTokenizer


@pytest.fixture(scope="module")
def client():
    init_parallel(1)
    cfg = EngineConfig(model=get_model_config("tiny-llama-test"), device="cpu",
                       max_num_seqs=8, num_gpu_blocks=128, enforce_eager=True,
                       max_model_len=128)
    eng = LLMEngine(cfg)
    aeng = AsyncLLMEngine(eng).start()
    app = build_app(aeng, ByteTokenizer(cfg.model.vocab_size), "tiny-llama-test")
    with TestClient(app) as c:
        yield c
    aeng.shutdown()


def test_health(client):
    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"


def test_models(client):
    r = client.get("/v1/models")
    assert r.json()["data"][0]["id"] == "tiny-llama-test"


def test_metrics_names(client):
    r = client.get("/metrics")
    body = r.text
    assert "vllm:cache_config_info" in body
    assert "vllm:num_requests_running" in body
    assert "vllm:generation_tokens" in body
    assert 'num_gpu_blocks="65"' in body  # capped: max_num_seqs*blocks/seq+1


def test_completions_tokens(client):
    r = client.post("/v1/completions", json={
        "prompt": [3, 14, 15, 92], "max_tokens": 6, "temperature": 0,
        "ignore_eos": True})
    assert r.status_code == 200
    data = r.json()
    assert data["object"] == "text_completion"
    assert data["usage"]["completion_tokens"] == 6
    assert data["choices"][0]["finish_reason"] == "length"


def test_completions_text_roundtrip(client):
    r = client.post("/v1/completions", json={
        "prompt": "hello", "max_tokens": 3, "temperature": 0,
        "ignore_eos": True})
    assert r.status_code == 200
    assert r.json()["usage"]["prompt_tokens"] == 6  # bos + 5 bytes


def test_chat_completion(client):
    r = client.post("/v1/chat/completions", json={
        "messages": [{"role": "user", "content": "hi"}],
        "max_tokens": 4, "temperature": 0, "ignore_eos": True})
    assert r.status_code == 200
    data = r.json()
    assert data["object"] == "chat.completion"
    assert data["choices"][0]["message"]["role"] == "assistant"


def test_streaming_sse(client):
    with client.stream("POST", "/v1/completions", json={
            "prompt": [5, 6, 7], "max_tokens": 4, "stream": True,
            "temperature": 0, "ignore_eos": True}) as r:
        assert r.status_code == 200
        lines = [ln for ln in r.iter_lines() if ln.startswith("data:")]
    assert lines[-1] == "data: [DONE]"
    assert len(lines) >= 5  # 4 tokens + finish + DONE


def test_rate_limit_429():
    """A saturated queue must 429 on generation endpoints but keep /health."""
    init_parallel(1)
    cfg = EngineConfig(model=get_model_config("tiny-llama-test"), device="cpu",
                       max_num_seqs=2, num_gpu_blocks=64, enforce_eager=True,
                       max_model_len=64)
    eng = LLMEngine(cfg)
    aeng = AsyncLLMEngine(eng)  # NOT started: queue only grows
    app = build_app(aeng, ByteTokenizer(cfg.model.vocab_size), "t",
                    max_queue=2)
    with TestClient(app) as c:
        # stuff the submit queue directly (loop not running)
        import asyncio
        from kaito_amd.engine.sequence import SamplingParams
        from kaito_amd.server.async_engine import _Pending
        loop = asyncio.new_event_loop()
        for _ in range(3):
            aeng._submit.put(_Pending([1, 2], SamplingParams(), None, loop))
        r = c.post("/v1/completions", json={"prompt": [1], "max_tokens": 1})
        assert r.status_code == 429
        assert c.get("/health").status_code == 200
        m = c.get("/metrics").text
        assert "kaito_ratelimit_rejected" in m


def test_kv_event_bus_roundtrip():
    from kaito_amd.engine.kv_events import (KVEventPublisher,
                                            KVEventSubscriber, BLOCK_STORED)
    pub = KVEventPublisher(host="127.0.0.1", port=0)
    sub = KVEventSubscriber(port=pub.port)
    deadline = time.monotonic() + 3
    while pub._pub.num_subscriptions < 1 and time.monotonic() < deadline:
        time.sleep(0.02)  # ZMQ slow-joiner: wait for the subscription
    pub.block_stored([1, 2, 3])
    pub.block_removed([2])
    pub.all_cleared()
    deadline = time.monotonic() + 3
    while len(sub.events) < 3 and time.monotonic() < deadline:
        time.sleep(0.05)
    assert len(sub.events) == 3
    assert sub.events[0]["event"] == BLOCK_STORED
    assert sub.events[0]["block_hashes"] == [1, 2, 3]
    assert sub.events[2]["event"] == "AllBlocksCleared"
    sub.close()
    pub.close()


def test_engine_publishes_kv_events():
    from kaito_amd.engine.kv_events import KVEventPublisher, KVEventSubscriber
    from kaito_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from kaito_amd.models import get_model_config
    init_parallel(1)
    cfg = EngineConfig(model=get_model_config("tiny-llama-test"), device="cpu",
                       max_num_seqs=4, num_gpu_blocks=64, enforce_eager=True,
                       max_model_len=64)
    eng = LLMEngine(cfg)
    pub = KVEventPublisher(host="127.0.0.1", port=0)
    eng.kv_publisher = pub
    sub = KVEventSubscriber(port=pub.port)
    deadline = time.monotonic() + 3
    while pub._pub.num_subscriptions < 1 and time.monotonic() < deadline:
        time.sleep(0.02)
    eng.generate([[1, 2, 3, 4]], SamplingParams(max_tokens=3, ignore_eos=True))
    deadline = time.monotonic() + 3
    while len(sub.events) < 2 and time.monotonic() < deadline:
        time.sleep(0.05)
    kinds = [e["event"] for e in sub.events]
    assert "BlockStored" in kinds and "BlockRemoved" in kinds
    sub.close()
    pub.close()


def test_entrypoint_config_file_merge(tmp_path, monkeypatch):
    """--kaito-config-file YAML merge: file overrides defaults, explicit
    CLI flags override the file (reference inference_api.py:128-154)."""
    import sys
    from kaito_amd.server.entrypoint import build_parser, merge_config_file
    cfg = tmp_path / "inference_config.yaml"
    cfg.write_text("""
vllm:
  max-num-seqs: 64
  gpu-memory-utilization: 0.8
  max-model-len: 2048
""")
    monkeypatch.setattr(sys, "argv",
                        ["prog", "--model", "llama-3-8b",
                         "--max-num-seqs", "128"])
    args = build_parser().parse_args(
        ["--model", "llama-3-8b", "--max-num-seqs", "128"])
    merge_config_file(args, str(cfg))
    assert args.max_num_seqs == 128        # CLI wins
    assert args.gpu_memory_utilization == 0.8   # file wins over default
    assert str(args.max_model_len) == "2048"


def test_transformers_fallback_runtime():
    """Fallback runtime serves phi-3-mini-class models (head_dim=96 — outside
    the native decode kernel's support matrix)."""
    from kaito_amd.server.transformers_runtime import (FallbackGenerator,
                                                       build_fallback_app)
    gen = FallbackGenerator("tiny-llama-test", device="cpu")
    app = build_fallback_app(gen, "tiny-llama-test")
    with TestClient(app) as c:
        assert c.get("/health").json()["status"] == "ok"
        assert c.get("/v1/models").json()["data"][0]["runtime"] == \
            "transformers"
        r = c.post("/v1/completions", json={"prompt": [4, 5, 6],
                                            "max_tokens": 3,
                                            "temperature": 0})
        assert r.status_code == 200
        assert r.json()["usage"]["completion_tokens"] == 3


def test_stop_strings_truncate(client):
    """'stop' strings cut the completion and abort the sequence."""
    # byte tokenizer: generated ids decode to bytes; pick the first
    # generated char as the stop string
    r0 = client.post("/v1/completions", json={
        "prompt": "ab", "max_tokens": 6, "temperature": 0,
        "ignore_eos": True})
    full = r0.json()["choices"][0]["text"]
    assert len(full) > 1
    stop = full[1]
    r = client.post("/v1/completions", json={
        "prompt": "ab", "max_tokens": 6, "temperature": 0,
        "ignore_eos": True, "stop": [stop]})
    data = r.json()
    assert data["choices"][0]["finish_reason"] == "stop"
    assert stop not in data["choices"][0]["text"]
    assert data["choices"][0]["text"] == full.split(stop)[0]


def test_completions_logprobs_and_penalties(client):
    c = client
    if True:
        r = c.post("/v1/completions", json={
            "prompt": [3, 14, 15], "max_tokens": 4, "temperature": 0.0,
            "ignore_eos": True, "logprobs": 2})
        assert r.status_code == 200
        lp = r.json()["choices"][0]["logprobs"]
        assert len(lp["tokens"]) == 4
        assert len(lp["token_logprobs"]) == 4
        assert all(v is not None and v <= 0 for v in lp["token_logprobs"])
        assert all(len(t) == 2 for t in lp["top_logprobs"])
        r2 = c.post("/v1/completions", json={
            "prompt": [3, 14, 15], "max_tokens": 6, "temperature": 0.0,
            "ignore_eos": True, "frequency_penalty": 50.0})
        assert r2.status_code == 200
        r3 = c.post("/v1/chat/completions", json={
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 3, "temperature": 0, "ignore_eos": True,
            "logprobs": True, "top_logprobs": 2})
        assert r3.status_code == 200
        content = r3.json()["choices"][0]["logprobs"]["content"]
        assert len(content) == 3
        assert all(len(e["top_logprobs"]) == 2 for e in content)


def test_completions_n_choices(client):
    r = client.post("/v1/completions", json={
        "prompt": [3, 14, 15], "max_tokens": 5, "temperature": 1.0,
        "seed": 7, "ignore_eos": True, "n": 3})
    assert r.status_code == 200
    ch = r.json()["choices"]
    assert [c["index"] for c in ch] == [0, 1, 2]
    assert all(c["text"] for c in ch)
    # seeded n-choices: per-index derived seeds → deterministic across calls
    r2 = client.post("/v1/completions", json={
        "prompt": [3, 14, 15], "max_tokens": 5, "temperature": 1.0,
        "seed": 7, "ignore_eos": True, "n": 3})
    assert [c["text"] for c in r2.json()["choices"]] == \
        [c["text"] for c in ch]
    assert r.json()["usage"]["completion_tokens"] == 15


def test_download_monitor_progress(tmp_path):
    """Bytes-on-disk sampling against the safetensors index total
    (reference: download-progress gauges, inference_api.py:265-365)."""
    import json
    from kaito_amd.server.download_monitor import (DownloadMonitor,
                                                   bytes_on_disk,
                                                   expected_total_bytes)
    d = tmp_path / "w"
    d.mkdir()
    (d / "model.safetensors.index.json").write_text(
        json.dumps({"metadata": {"total_size": 1000}}))
    assert expected_total_bytes(str(d)) == 1000
    mon = DownloadMonitor(str(d), interval_s=0.01)
    assert mon.progress() == 0.0
    (d / "a.safetensors").write_bytes(b"x" * 400)
    assert bytes_on_disk(str(d)) == 400
    assert mon.progress() == 0.4
    (d / "b.safetensors").write_bytes(b"y" * 900)   # over-report clamps
    assert mon.progress() == 1.0
    mon.start()
    time.sleep(0.05)
    mon.stop(done=True)
    from kaito_amd.server import metrics as M
    assert M.MODEL_DOWNLOAD_DONE._value.get() == 1


def test_gpu_cache_usage_metric(client):
    client.post("/v1/completions", json={"prompt": [3, 4, 5],
                                         "max_tokens": 2,
                                         "ignore_eos": True})
    m = client.get("/metrics").text
    line = [ln for ln in m.splitlines()
            if ln.startswith("vllm:gpu_cache_usage_perc ")]
    assert line and 0.0 <= float(line[0].split()[-1]) <= 1.0


def test_ttft_tpot_histograms(client):
    client.post("/v1/completions", json={"prompt": [9, 8, 7],
                                         "max_tokens": 4,
                                         "ignore_eos": True})
    m = client.get("/metrics").text
    assert "vllm:time_to_first_token_seconds_bucket" in m
    assert "vllm:time_per_output_token_seconds_bucket" in m
    count = [ln for ln in m.splitlines()
             if ln.startswith("vllm:time_to_first_token_seconds_count")]
    assert count and float(count[0].split()[-1]) >= 1
