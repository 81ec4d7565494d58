"""Operator entrypoint — the analog of the reference's cmd/workspace/main.go
(flag parsing :100-146, feature gates :154, SKU handler init :159-164,
provisioner selection :257-286, controller registration :300-385).

Runs the reconcile loop against a KubeClient. The in-cluster client wraps
the `kubernetes` package when installed; otherwise (dev/test) the fake
client drives everything in-memory.
"""
from __future__ import annotations

import argparse
import logging
import time
from typing import Dict, Optional

from .api_types import (Workspace, ResourceSpec, InferenceSpec, TuningSpec,
                        PresetSpec, DataSource, DataDestination,
                        InferenceSet, InferenceSetSpec)
from .controllers.inferenceset import InferenceSetReconciler
from .controllers.workspace import WorkspaceReconciler
from .featuregates import parse_feature_gates
from .kubeclient import FakeKubeClient, KubeClient
from .nodeprovision import make_provisioner
from .sku import get_sku_handler

logger = logging.getLogger("kaito_amd.operator")


def workspace_from_obj(obj: Dict) -> Workspace:
    """Deserialize a stored Workspace object dict into the typed API."""
    spec = obj.get("spec", {})
    res = spec.get("resource", {})
    ws = Workspace(
        name=obj["metadata"]["name"],
        namespace=obj["metadata"].get("namespace", "default"),
        resource=ResourceSpec(
            instanceType=res.get("instanceType", ""),
            labelSelector=res.get("labelSelector", {}) or {},
            preferredNodes=res.get("preferredNodes", []) or [],
            count=res.get("count")),
        annotations=obj["metadata"].get("annotations", {}) or {},
        labels=obj["metadata"].get("labels", {}) or {},
        deletionTimestamp=obj["metadata"].get("deletionTimestamp"),
        finalizers=obj["metadata"].get("finalizers", []) or [],
    )
    inf = spec.get("inference")
    if inf:
        preset = inf.get("preset")
        name = preset.get("name") if isinstance(preset, dict) else preset
        ws.inference = InferenceSpec(
            preset=PresetSpec(name=name) if name else None,
            template=inf.get("template"),
            config=inf.get("config", ""))
    tun = spec.get("tuning")
    if tun:
        preset = tun.get("preset")
        name = preset.get("name") if isinstance(preset, dict) else preset
        ws.tuning = TuningSpec(
            preset=PresetSpec(name=name) if name else None,
            method=tun.get("method", "lora"),
            input=DataSource(**(tun.get("input") or {})),
            output=DataDestination(**(tun.get("output") or {})))
    return ws


def inferenceset_from_obj(obj: Dict) -> InferenceSet:
    spec = obj.get("spec", {})
    tpl_obj = {"metadata": {"name": obj["metadata"]["name"] + "-tpl",
                            "namespace": obj["metadata"].get("namespace",
                                                             "default")},
               "spec": spec.get("workspaceTemplate", {})}
    return InferenceSet(
        name=obj["metadata"]["name"],
        namespace=obj["metadata"].get("namespace", "default"),
        deletionTimestamp=obj["metadata"].get("deletionTimestamp"),
        finalizers=obj["metadata"].get("finalizers", []) or [],
        spec=InferenceSetSpec(
            replicas=spec.get("replicas", 1),
            workspaceTemplate=workspace_from_obj(tpl_obj),
            upgradeStrategy=spec.get("upgradeStrategy", "Surge"),
            maintenanceWindow=spec.get("maintenanceWindow", "")))


class OperatorLoop:
    """Polling reconcile driver (controller-runtime informer analog)."""

    def __init__(self, client: KubeClient, cloud: str = "azure",
                 provisioner: str = "byo", image: str = "ghcr.io/kaito-amd/engine:latest",
                 gates: Optional[Dict[str, bool]] = None):
        self.client = client
        self.gates = gates or {}
        sku = get_sku_handler(cloud)
        prov = make_provisioner(provisioner, client)
        self.workspace = WorkspaceReconciler(client, sku, prov, image)
        self.inferenceset = InferenceSetReconciler(client) \
            if self.gates.get("enableInferenceSetController", True) else None

    def tick(self) -> int:
        """One reconcile pass over all stored CRs. Returns CR count."""
        n = 0
        if self.inferenceset is not None:
            for obj in self.client.list("InferenceSet"):
                try:
                    self.inferenceset.reconcile(inferenceset_from_obj(obj))
                except Exception:  # noqa: BLE001
                    logger.exception("inferenceset %s reconcile failed",
                                     obj["metadata"]["name"])
                n += 1
        for obj in self.client.list("Workspace"):
            ws = workspace_from_obj(obj)
            try:
                self.workspace.reconcile(ws)
            except Exception:  # noqa: BLE001
                logger.exception("workspace %s reconcile failed", ws.name)
            n += 1
        from .metrics import monitor_workspaces
        monitor_workspaces(self.client.list("Workspace"))
        return n

    def run(self, interval_s: float = 5.0, max_ticks: Optional[int] = None):
        t = 0
        while max_ticks is None or t < max_ticks:
            self.tick()
            t += 1
            time.sleep(interval_s)


def main(argv=None):
    logging.basicConfig(level=logging.INFO)
    p = argparse.ArgumentParser()
    p.add_argument("--feature-gates", default="")
    p.add_argument("--cloud-provider", default="azure")
    p.add_argument("--node-provisioner", default="karpenter")
    p.add_argument("--preset-image",
                   default="ghcr.io/kaito-amd/engine:latest")
    p.add_argument("--reconcile-interval", type=float, default=5.0)
    args = p.parse_args(argv)
    gates = parse_feature_gates(args.feature_gates)
    from .kubeclient_incluster import make_kube_client
    client = make_kube_client()
    logger.info("kube client: %s", type(client).__name__)
    loop = OperatorLoop(client, args.cloud_provider, args.node_provisioner,
                        args.preset_image, gates)
    loop.run(args.reconcile_interval)


if __name__ == "__main__":
    main()
