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
    """Deserialize a stored Workspace object dict into the typed API.

    The reference embeds resource/inference/tuning at the TOP level of
    the Workspace object (workspace_types.go:298-306 — no .spec
    wrapper), which the CRD schema mirrors; a legacy .spec wrapper is
    still accepted."""
    spec = obj if ("resource" in obj or "inference" in obj
                   or "tuning" in obj) else obj.get("spec", {})
    res = spec.get("resource", {})
    part = res.get("partition") or None
    from .api_types import AdapterSpec, PartitionSpec
    ws = Workspace(
        name=obj["metadata"]["name"],
        namespace=obj["metadata"].get("namespace", "default"),
        resource=ResourceSpec(
            instanceType=res.get("instanceType", ""),
            labelSelector=res.get("labelSelector", {}) or {},
            preferredNodes=res.get("preferredNodes", []) or [],
            count=res.get("count"),
            partition=PartitionSpec(
                partitionType=part.get("partitionType"),
                partitionCount=part.get("partitionCount"))
            if part else None),
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
            config=inf.get("config", ""),
            adapters=[AdapterSpec(source=a.get("source", {}) or {},
                                  strength=a.get("strength"))
                      for a in inf.get("adapters", []) or []])
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


def ragengine_from_obj(obj: Dict):
    from .api_types import RAGEngine, RAGEngineSpec
    spec = obj.get("spec", {})
    return RAGEngine(
        name=obj["metadata"]["name"],
        namespace=obj["metadata"].get("namespace", "default"),
        deletionTimestamp=obj["metadata"].get("deletionTimestamp"),
        spec=RAGEngineSpec(
            embedding=spec.get("embedding", {}) or {},
            inferenceService=spec.get("inferenceService", {}) or {},
            storage=spec.get("storage", {}) or {},
            guardrails=spec.get("guardrails", {}) or {},
            indexServiceName=spec.get("indexServiceName", ""),
            queryServiceName=spec.get("queryServiceName", "")))


def multirole_from_obj(obj: Dict):
    from .controllers.multirole import MultiRoleInference, RoleSpec
    spec = obj.get("spec", {})
    return MultiRoleInference(
        name=obj["metadata"]["name"],
        namespace=obj["metadata"].get("namespace", "default"),
        preset=spec.get("preset", ""),
        prefill=RoleSpec(**(spec.get("prefill") or {})),
        decode=RoleSpec(**(spec.get("decode") or {})))


def modelmirror_from_obj(obj: Dict):
    from .controllers.modelmirror import ModelMirror
    spec = obj.get("spec", {})
    return ModelMirror(
        name=obj["metadata"]["name"],
        modelName=spec.get("modelName", ""),
        mode=spec.get("mode", "Managed"),
        storageClassName=spec.get("storageClassName",
                                  "kaito-local-nvme-disk"),
        storageSize=spec.get("storageSize", "200Gi"),
        staticVolumePath=spec.get("staticVolumePath", ""),
        namespace=spec.get("namespace", "kaito-system"))


class OperatorLoop:
    """Polling reconcile driver (controller-runtime informer analog)."""

    def __init__(self, client: KubeClient, cloud: str = "azure",
                 provisioner: str = "byo", image: str = "ghcr.io/kaito-amd/engine:latest",
                 gates: Optional[Dict[str, bool]] = None,
                 controllers: str = "all"):
        self.client = client
        self.gates = gates or {}
        only = set(controllers.split(",")) if controllers != "all" else None

        def want(name: str, gate: Optional[str] = None,
                 default: bool = True) -> bool:
            if only is not None:
                return name in only
            return self.gates.get(gate, default) if gate else True

        sku = get_sku_handler(cloud)
        prov = make_provisioner(provisioner, client)
        self.workspace = WorkspaceReconciler(client, sku, prov, image) \
            if want("workspace") else None
        self.inferenceset = InferenceSetReconciler(client) \
            if want("inferenceset", "enableInferenceSetController") else None
        self.ragengine = None
        if want("ragengine"):
            from .controllers.ragengine import RAGEngineReconciler
            self.ragengine = RAGEngineReconciler(client)
        self.multirole = None
        if want("multirole", "enableMultiRoleInferenceController", False):
            from .controllers.multirole import MultiRoleInferenceReconciler
            self.multirole = MultiRoleInferenceReconciler(client)
        self.modelmirror = None
        if want("modelmirror", "ModelMirror", False):
            from .controllers.modelmirror import ModelMirrorReconciler
            self.modelmirror = ModelMirrorReconciler(client)

    def _each(self, kind: str, reconciler, convert) -> int:
        n = 0
        if reconciler is None:
            return 0
        for obj in self.client.list(kind):
            try:
                reconciler.reconcile(convert(obj))
            except Exception:  # noqa: BLE001
                logger.exception("%s %s reconcile failed", kind,
                                 obj["metadata"]["name"])
            n += 1
        return n

    def tick(self) -> int:
        """One reconcile pass over all stored CRs. Returns CR count."""
        n = 0
        n += self._each("ModelMirror", self.modelmirror, modelmirror_from_obj)
        n += self._each("MultiRoleInference", self.multirole,
                        multirole_from_obj)
        n += self._each("InferenceSet", self.inferenceset,
                        inferenceset_from_obj)
        n += self._each("Workspace", self.workspace, workspace_from_obj)
        n += self._each("RAGEngine", self.ragengine, ragengine_from_obj)
        if self.workspace is not None:
            from .metrics import monitor_workspaces
            monitor_workspaces(self.client.list("Workspace"))
        return n

    def run(self, interval_s: float = 5.0, max_ticks: Optional[int] = None,
            elector=None):
        """Reconcile loop; with an elector, only the Lease holder
        reconciles (controller-runtime leader-election semantics —
        standby replicas keep renewing their candidacy)."""
        t = 0
        while max_ticks is None or t < max_ticks:
            if elector is None or elector.try_acquire():
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
    p.add_argument("--controllers", default="all",
                   help="comma list (workspace,inferenceset,ragengine,"
                        "multirole,modelmirror) or 'all' (feature-gated)")
    p.add_argument("--leader-elect", action="store_true", default=True)
    p.add_argument("--no-leader-elect", dest="leader_elect",
                   action="store_false")
    p.add_argument("--bootstrap-webhook-cert", action="store_true",
                   help="generate a self-signed serving cert and patch "
                        "the ValidatingWebhookConfiguration caBundle")
    args = p.parse_args(argv)
    gates = parse_feature_gates(args.feature_gates)
    from .kubeclient_incluster import make_kube_client
    client = make_kube_client()
    logger.info("kube client: %s", type(client).__name__)
    if args.bootstrap_webhook_cert:
        from .leader import generate_self_signed_cert, patch_webhook_ca_bundle
        crt, key, ca = generate_self_signed_cert()
        patched = patch_webhook_ca_bundle(client, ca)
        logger.info("webhook cert bootstrap: cert=%s patched=%s", crt,
                    patched)
    elector = None
    if args.leader_elect:
        from .leader import LeaderElector
        elector = LeaderElector(client)
    loop = OperatorLoop(client, args.cloud_provider, args.node_provisioner,
                        args.preset_image, gates,
                        controllers=args.controllers)
    loop.run(args.reconcile_interval, elector=elector)


if __name__ == "__main__":
    main()
