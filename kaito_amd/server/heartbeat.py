"""Per-rank liveness heartbeats — the torchrun-era replacement for the
reference's Ray actor-death scan (multi-node-health-check.py:21-50).

Every engine rank writes its OWN heartbeat file containing its pid and
a timestamp from inside the step loop; the liveness probe then requires
EVERY expected local rank to be (a) fresh and (b) backed by a live
process. A hung rank stops refreshing its own file and no other process
can mask it (the round-1 single shared-mtime heartbeat could be kept
alive by any survivor)."""
from __future__ import annotations

import os
import time
from typing import List, Optional, Tuple

def _default_dir() -> str:
    return os.environ.get("KAITO_HEARTBEAT_DIR", "/tmp/kaito_heartbeats")


class Heartbeat:
    """Writer side: call beat() from the engine loop (throttled)."""

    def __init__(self, rank: int, directory: Optional[str] = None,
                 interval_s: float = 1.0):
        directory = directory or _default_dir()
        self.rank = rank
        self.dir = directory
        self.interval_s = interval_s
        self._last = 0.0
        os.makedirs(directory, exist_ok=True)
        self.path = os.path.join(directory, f"rank{rank}")
        self.beat(force=True)

    def beat(self, force: bool = False) -> None:
        now = time.time()
        if not force and now - self._last < self.interval_s:
            return
        self._last = now
        tmp = self.path + ".tmp"
        with open(tmp, "w") as f:
            f.write(f"{os.getpid()} {now}\n")
        os.replace(tmp, self.path)


def _pid_alive(pid: int) -> bool:
    try:
        os.kill(pid, 0)
        return True
    except ProcessLookupError:
        return False
    except PermissionError:
        return True


def check_all(expected_ranks: int, directory: Optional[str] = None,
              max_age_s: float = 120.0,
              ranks: Optional[List[int]] = None) -> Tuple[bool, str]:
    """Liveness: every expected rank fresh AND its pid alive."""
    directory = directory or _default_dir()
    now = time.time()
    for r in (ranks if ranks is not None else range(expected_ranks)):
        path = os.path.join(directory, f"rank{r}")
        try:
            with open(path) as f:
                pid_s, ts_s = f.read().split()
        except (OSError, ValueError):
            return False, f"rank {r}: no heartbeat"
        if now - float(ts_s) > max_age_s:
            return False, f"rank {r}: stale ({now - float(ts_s):.0f}s)"
        if not _pid_alive(int(pid_s)):
            return False, f"rank {r}: pid {pid_s} dead"
    return True, "ok"
