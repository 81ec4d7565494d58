"""Controller Prometheus metrics — parity with the reference's
pkg/workspace/controllers/metrics.go:31-60 (kaito_workspace_count{phase},
preset counts) and pkg/inferenceset/metrics.go."""
from __future__ import annotations

from collections import Counter
from typing import List

from prometheus_client import CollectorRegistry, Gauge, generate_latest

OPERATOR_REGISTRY = CollectorRegistry()

WORKSPACE_COUNT = Gauge("kaito_workspace_count", "Workspaces by phase",
                        ["phase"], registry=OPERATOR_REGISTRY)
WORKSPACE_PRESET_COUNT = Gauge("kaito_workspace_preset_count",
                               "Workspaces by preset", ["preset"],
                               registry=OPERATOR_REGISTRY)
INFERENCESET_COUNT = Gauge("kaito_inferenceset_count",
                           "InferenceSets by ready state", ["ready"],
                           registry=OPERATOR_REGISTRY)


def monitor_workspaces(workspaces: List[dict]) -> None:
    """Poller body (reference: monitorWorkspaces goroutine)."""
    phases = Counter(w.get("status", {}).get("state", "Pending")
                     for w in workspaces)
    for phase in ("Pending", "Ready", "NotReady", "Running", "Succeeded",
                  "Failed"):
        WORKSPACE_COUNT.labels(phase=phase).set(phases.get(phase, 0))
    presets = Counter()
    for w in workspaces:
        p = (w.get("spec", {}).get("inference", {}) or {}).get("preset")
        name = p.get("name") if isinstance(p, dict) else p
        if name:
            presets[name] += 1
    for preset, n in presets.items():
        WORKSPACE_PRESET_COUNT.labels(preset=preset).set(n)


def render() -> bytes:
    return generate_latest(OPERATOR_REGISTRY)
