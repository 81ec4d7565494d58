"""Multi-node liveness/readiness probes — parity with the reference's
presets/workspace/inference/vllm/multi-node-health-check.py:21-50.

Readiness: the leader's /health.
Liveness: PER-RANK heartbeats (kaito_amd.server.heartbeat): every local
engine rank must have refreshed its own file recently AND its recorded
pid must be alive — the analog of the reference's Ray actor-death scan.
A hung rank fails its own heartbeat; a dead rank fails the pid check;
no surviving process can mask either (the round-1 shared-mtime file
could)."""
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
    else:  # liveness
        from kaito_amd.server.heartbeat import check_all
        n = int(os.environ.get("LOCAL_WORLD_SIZE",
                               os.environ.get("KAITO_LOCAL_RANKS", "1")))
        node = int(os.environ.get("GROUP_RANK",
                                  os.environ.get("POD_INDEX", "0") or "0"))
        ranks = [node * n + i for i in range(n)]   # this node's GLOBAL ranks
        ok, detail = check_all(n, ranks=ranks)
        if not ok:
            print(f"liveness: {detail}", file=sys.stderr)
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
