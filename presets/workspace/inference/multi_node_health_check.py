"""Multi-node liveness/readiness probes — parity with the reference's
presets/workspace/inference/vllm/multi-node-health-check.py:21-50:
liveness fails when the distributed worker group lost a member (our
torchrun-based bootstrap: the rendezvous store heartbeat); readiness is
the leader's /health."""
import os
import sys
import urllib.request


def leader_health(host: str, port: int = 5000) -> bool:
    try:
        with urllib.request.urlopen(f"http://{host}:{port}/health",
                                    timeout=5) as r:
            return r.status == 200
    except OSError:
        return False


def main():
    mode = sys.argv[1] if len(sys.argv) > 1 else "readiness"
    leader = os.environ.get("KAITO_LEADER_HOST", "127.0.0.1")
    if mode == "readiness":
        ok = leader_health(leader)
    else:  # liveness: engine process heartbeat file updated by the runner
        hb = os.environ.get("KAITO_HEARTBEAT_FILE", "/tmp/kaito_heartbeat")
        import time
        try:
            ok = time.time() - os.path.getmtime(hb) < 120
        except OSError:
            # no heartbeat file yet: fall back to leader health
            ok = leader_health(leader)
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
