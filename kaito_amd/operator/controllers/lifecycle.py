"""Drift remediation + base-image auto-upgrade.

Drift (reference pkg/controllers/drift/controller.go:35-49): Karpenter-mode
rolling node replacement — at most ONE workspace's nodes drift-remediated
at a time per InferenceSet; a disruption budget gate opens (1) for the
workspace under remediation and stays closed (0) for the rest.

Auto-upgrade (reference pkg/controllers/autoupgrade/runner.go:50+): a
background poller that detects base-image drift on InferenceSet-owned
workspaces and applies the InPlace or Surge strategy inside a cron-windowed
maintenance window (api/v1beta1/inferenceset_types.go:48-110).
"""
from __future__ import annotations

import time
from typing import Dict, List, Optional

from ..api_types import (LABEL_INFERENCESET_CREATED_BY,
                         LABEL_UPGRADE_TO_VERSION)
from ..kubeclient import KubeClient, NotFound


# --------------------------------------------------------------------- drift
class DriftReconciler:
    def __init__(self, client: KubeClient):
        self.client = client
        # inferenceset name → workspace currently being remediated
        self._active: Dict[str, str] = {}

    def reconcile(self, iset_name: str, namespace: str = "default"
                  ) -> Optional[str]:
        """Serializes drift across an InferenceSet's workspaces: returns the
        workspace under remediation (requeue while active) or None."""
        children = self.client.list("Workspace", namespace, {
            LABEL_INFERENCESET_CREATED_BY: iset_name})
        drifted = [c for c in children
                   if c.get("metadata", {}).get("annotations", {})
                   .get("kaito.sh/node-drifted") == "true"]
        active = self._active.get(iset_name)
        if active:
            # remediation in progress: done when the annotation clears
            still = any(c["metadata"]["name"] == active for c in drifted)
            if still:
                return active
            self._set_budget(namespace, active, 0)
            del self._active[iset_name]
        if not drifted:
            return None
        victim = drifted[0]["metadata"]["name"]
        self._active[iset_name] = victim
        self._set_budget(namespace, victim, 1)
        return victim

    def _set_budget(self, namespace: str, ws_name: str, budget: int):
        """Toggle the workspace's NodePool disruption budget 0↔1
        (reference :35-39)."""
        name = f"{ws_name}-nodepool"
        try:
            np = self.client.get("NodePool", namespace, name)
        except NotFound:
            np = self.client.create({
                "apiVersion": "karpenter.sh/v1", "kind": "NodePool",
                "metadata": {"name": name, "namespace": namespace},
                "spec": {"disruption": {"budgets": [{"nodes": "0"}]}},
            })
        np["spec"]["disruption"]["budgets"] = [{"nodes": str(budget)}]
        self.client.update(np)


# -------------------------------------------------------------- auto-upgrade
def in_maintenance_window(cron: str, now: Optional[time.struct_time] = None
                          ) -> bool:
    """Minimal cron-window check: 'M H * * D' fields; '*' matches. Window =
    the hour starting at the cron time."""
    if not cron:
        return True
    now = now or time.localtime()
    parts = cron.split()
    if len(parts) != 5:
        return True
    _minute, hour, _dom, _month, dow = parts

    def match(spec, value):
        if spec == "*":
            return True
        return any(int(x) == value for x in spec.split(",") if x.isdigit())

    return match(hour, now.tm_hour) and match(dow, now.tm_wday)


class AutoUpgradeRunner:
    """Applies the upgrade label to out-of-date workspaces; Surge lets the
    InferenceSet controller replace them old-first keeping Ready >= desired,
    InPlace updates the image directly."""

    def __init__(self, client: KubeClient, target_revision: str,
                 strategy: str = "Surge", maintenance_window: str = ""):
        self.client = client
        self.target = target_revision
        self.strategy = strategy
        self.window = maintenance_window

    def poll(self, iset_name: str, namespace: str = "default") -> List[str]:
        """One poll tick. Returns workspaces marked for upgrade."""
        if not in_maintenance_window(self.window):
            return []
        marked = []
        for ws in self.client.list("Workspace", namespace, {
                LABEL_INFERENCESET_CREATED_BY: iset_name}):
            labels = ws["metadata"].setdefault("labels", {})
            current = labels.get("inferenceset.kaito.io/revision")
            if current == self.target:
                continue
            if self.strategy == "InPlace":
                labels["inferenceset.kaito.io/revision"] = self.target
                self.client.update(ws)
            else:  # Surge: mark; the InferenceSet controller rolls old-first
                if labels.get(LABEL_UPGRADE_TO_VERSION) != self.target:
                    labels[LABEL_UPGRADE_TO_VERSION] = self.target
                    self.client.update(ws)
            marked.append(ws["metadata"]["name"])
        return marked
