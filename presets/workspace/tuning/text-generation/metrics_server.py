"""Tuning metrics sidecar — reference layout parity with
presets/workspace/tuning/text-generation/metrics/metrics_server.py
(GPU/CPU/memory gauges on :5000). Wraps kaito_amd.tuning.metrics_server."""
import sys

from kaito_amd.tuning.metrics_server import main

if __name__ == "__main__":
    sys.exit(main())
