"""Model download/progress monitor — parity with the reference's
download-progress machinery (presets/workspace/inference/vllm/
inference_api.py:48-61,265-365: prometheus gauges fed by bytes-on-disk
sampling while weights stream in; the workspace controller and the
pre-load /metrics server surface them).

watch() samples the weights directory's total byte size against the
expected total (safetensors index metadata when present) and updates
kaito_model_download_progress until the load completes or the expected
bytes arrive.
"""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Optional

from . import metrics


def expected_total_bytes(weights_path: str) -> Optional[int]:
    """Expected checkpoint size: safetensors index metadata, else None."""
    idx = os.path.join(weights_path, "model.safetensors.index.json")
    if os.path.exists(idx):
        try:
            with open(idx) as f:
                t = json.load(f).get("metadata", {}).get("total_size")
            return int(t) if t else None
        except (OSError, ValueError):
            return None
    return None


def bytes_on_disk(weights_path: str) -> int:
    total = 0
    for root, _, files in os.walk(weights_path):
        for fn in files:
            if fn.endswith((".safetensors", ".bin", ".pt")):
                try:
                    total += os.path.getsize(os.path.join(root, fn))
                except OSError:
                    pass
    return total


class DownloadMonitor:
    """Background sampler: progress = bytes_on_disk / expected (clamped).
    Without an expected total it reports 0 until stop(done=True)."""

    def __init__(self, weights_path: str, interval_s: float = 2.0):
        self.weights_path = weights_path
        self.interval_s = interval_s
        self.expected = expected_total_bytes(weights_path)
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def progress(self) -> float:
        if not self.expected:
            return 0.0
        return min(bytes_on_disk(self.weights_path) / self.expected, 1.0)

    def _loop(self):
        while not self._stop.is_set():
            metrics.MODEL_DOWNLOAD_PROGRESS.set(self.progress())
            self._stop.wait(self.interval_s)

    def start(self) -> "DownloadMonitor":
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self, done: bool = True) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
        if done:
            metrics.MODEL_DOWNLOAD_PROGRESS.set(1.0)
            metrics.MODEL_DOWNLOAD_DONE.set(1)
