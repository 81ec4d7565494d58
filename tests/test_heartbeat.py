"""Per-rank heartbeat liveness (the Ray actor-death-scan analog)."""
import os
import subprocess
import sys
import time

from kaito_amd.server.heartbeat import Heartbeat, check_all


def test_heartbeat_fresh_and_alive(tmp_path):
    d = str(tmp_path)
    Heartbeat(0, d)
    Heartbeat(1, d)
    ok, detail = check_all(2, d)
    assert ok, detail


def test_heartbeat_missing_rank(tmp_path):
    d = str(tmp_path)
    Heartbeat(0, d)
    ok, detail = check_all(2, d)
    assert not ok and "rank 1" in detail


def test_heartbeat_stale_rank_fails(tmp_path):
    d = str(tmp_path)
    hb = Heartbeat(0, d)
    with open(hb.path, "w") as f:
        f.write(f"{os.getpid()} {time.time() - 1000}\n")
    ok, detail = check_all(1, d)
    assert not ok and "stale" in detail


def test_heartbeat_dead_pid_fails(tmp_path):
    """A rank whose process DIED fails liveness even with a fresh file —
    the failure mode the shared-mtime heartbeat could not see."""
    d = str(tmp_path)
    p = subprocess.Popen([sys.executable, "-c", "pass"])
    p.wait()
    with open(os.path.join(d, "rank0"), "w") as f:
        f.write(f"{p.pid} {time.time()}\n")
    ok, detail = check_all(1, d)
    assert not ok and "dead" in detail


def test_heartbeat_explicit_global_ranks(tmp_path):
    """Node 1 of a 2-node world checks ITS global ranks (8..15-style)."""
    d = str(tmp_path)
    Heartbeat(2, d)
    Heartbeat(3, d)
    ok, _ = check_all(2, d, ranks=[2, 3])
    assert ok
    ok, detail = check_all(2, d, ranks=[0, 1])
    assert not ok


def test_async_engine_writes_heartbeat(tmp_path, monkeypatch):
    monkeypatch.setenv("KAITO_HEARTBEAT_DIR", str(tmp_path))
    monkeypatch.setenv("RANK", "0")
    from kaito_amd.engine import EngineConfig, LLMEngine
    from kaito_amd.models import get_model_config
    from kaito_amd.parallel.state import init_parallel
    from kaito_amd.server.async_engine import AsyncLLMEngine
    init_parallel(1)
    eng = LLMEngine(EngineConfig(model=get_model_config("tiny-llama-test"),
                                 device="cpu", max_num_seqs=4,
                                 num_gpu_blocks=64, enforce_eager=True,
                                 max_model_len=64))
    a = AsyncLLMEngine(eng).start()
    try:
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline:
            ok, _ = check_all(1, str(tmp_path))
            if ok:
                break
            time.sleep(0.05)
        assert ok
    finally:
        a.shutdown()
