"""Node provisioning — rewrite of the reference's pkg/nodeprovision
(provisioner.go:36-100 NodeProvisioner interface; karpenter/provisioner.go;
byo-provisioner/byo_provisioner.go).

Implementations:
  * BYOProvisioner — no-op provisioning; selects pre-existing nodes by
    label selector / preferred nodes (reference byo_provisioner.go:136L).
  * KarpenterProvisioner — creates NodeClaim objects (cloud-agnostic CRs);
    readiness = NodeClaim.status.conditions Ready + a matching Node object.
"""
from __future__ import annotations

from typing import Dict, List

from .api_types import LABEL_WORKSPACE_NAME, LABEL_WORKSPACE_NAMESPACE, Workspace
from .kubeclient import KubeClient, NotFound


class NodeProvisioner:
    """Reference parity: NodeProvisioner (provisioner.go:36-100)."""

    def provision_nodes(self, ws: Workspace, count: int) -> List[str]:
        raise NotImplementedError

    def ensure_nodes_ready(self, ws: Workspace, count: int) -> List[str]:
        """Returns ready node names; fewer than `count` → requeue."""
        raise NotImplementedError

    def delete_nodes(self, ws: Workspace) -> None:
        raise NotImplementedError

    def build_node_selector(self, ws: Workspace) -> Dict[str, str]:
        sel = dict(ws.resource.labelSelector.get("matchLabels", {})) \
            if ws.resource.labelSelector else {}
        if ws.resource.instanceType:
            sel["node.kubernetes.io/instance-type"] = ws.resource.instanceType
        return sel


def _node_ready(node) -> bool:
    for c in node.get("status", {}).get("conditions", []):
        if c.get("type") == "Ready":
            return c.get("status") == "True"
    return False


class BYOProvisioner(NodeProvisioner):
    def __init__(self, client: KubeClient):
        self.client = client

    def provision_nodes(self, ws, count):
        return []  # bring-your-own: nothing to create

    def ensure_nodes_ready(self, ws, count):
        sel = self.build_node_selector(ws)
        nodes = self.client.list("Node", label_selector=sel or None)
        ready = [n["metadata"]["name"] for n in nodes if _node_ready(n)]
        preferred = [n for n in ws.resource.preferredNodes if n in ready]
        rest = [n for n in ready if n not in preferred]
        return (preferred + rest)[:count]

    def delete_nodes(self, ws):
        pass


class KarpenterProvisioner(NodeProvisioner):
    def __init__(self, client: KubeClient, node_class: str = "default"):
        self.client = client
        self.node_class = node_class

    def _claim_name(self, ws: Workspace, i: int) -> str:
        return f"{ws.name}-nc-{i}"

    def provision_nodes(self, ws, count):
        created = []
        for i in range(count):
            name = self._claim_name(ws, i)
            try:
                self.client.get("NodeClaim", ws.namespace, name)
            except NotFound:
                self.client.create({
                    "apiVersion": "karpenter.sh/v1",
                    "kind": "NodeClaim",
                    "metadata": {
                        "name": name, "namespace": ws.namespace,
                        "labels": {
                            LABEL_WORKSPACE_NAME: ws.name,
                            LABEL_WORKSPACE_NAMESPACE: ws.namespace,
                        },
                    },
                    "spec": {
                        "nodeClassRef": {"name": self.node_class},
                        "requirements": [{
                            "key": "node.kubernetes.io/instance-type",
                            "operator": "In",
                            "values": [ws.resource.instanceType],
                        }],
                        "resources": {"requests": {"amd.com/gpu": "1"}},
                    },
                    "status": {},
                })
                created.append(name)
        return created

    def ensure_nodes_ready(self, ws, count):
        ready = []
        claims = self.client.list("NodeClaim", ws.namespace, {
            LABEL_WORKSPACE_NAME: ws.name})
        for claim in claims:
            node_name = claim.get("status", {}).get("nodeName")
            if not node_name:
                continue
            try:
                node = self.client.get("Node", "", node_name)
            except NotFound:
                continue
            if _node_ready(node):
                ready.append(node_name)
        return ready[:count]

    def delete_nodes(self, ws):
        for claim in self.client.list("NodeClaim", ws.namespace, {
                LABEL_WORKSPACE_NAME: ws.name}):
            self.client.delete("NodeClaim", ws.namespace,
                               claim["metadata"]["name"])


    # ---- NodePool / NodeClass management (karpenter/nodepool.go) ----
    def ensure_node_pool(self, ws: Workspace, replicas: int = 1) -> Dict:
        """Per-workspace NodePool with a zero disruption budget (nodes
        are replaced only when the drift controller opens the budget —
        reference karpenter/provisioner.go + nodepool.go)."""
        name = f"{ws.name}-pool"
        try:
            return self.client.get("NodePool", ws.namespace, name)
        except NotFound:
            pool = {
                "apiVersion": "karpenter.sh/v1", "kind": "NodePool",
                "metadata": {"name": name, "namespace": ws.namespace,
                             "labels": {LABEL_WORKSPACE_NAME: ws.name}},
                "spec": {
                    "disruption": {"budgets": [{"nodes": "0"}],
                                   "consolidationPolicy": "WhenEmpty"},
                    "limits": {"amd.com/gpu": str(replicas * 8)},
                    "template": {"spec": {
                        "nodeClassRef": {"name": self.node_class},
                        "requirements": [{
                            "key": "node.kubernetes.io/instance-type",
                            "operator": "In",
                            "values": [ws.resource.instanceType],
                        }],
                    }},
                },
            }
            self.client.create(pool)
            return pool

    def set_drift_remediation(self, ws: Workspace, enabled: bool) -> None:
        """Open/close the NodePool disruption budget (drift controller)."""
        try:
            np = self.client.get("NodePool", ws.namespace, f"{ws.name}-pool")
        except NotFound:
            return
        np["spec"].setdefault("disruption", {})["budgets"] = [
            {"nodes": "1" if enabled else "0"}]
        self.client.update(np)


class AzureGPUProvisioner(NodeProvisioner):
    """azure-gpu-provisioner analog (reference
    gpu-provisioner/gpu_provisioner.go, 283 L): one NodeClaim per node
    with the Azure VM SKU requirement and kaito ownership labels; the
    azure machine controller fulfils the claim. Differs from the
    Karpenter path in claim shape (capacity-type on-demand, azure
    node-class group) and in that no NodePool is managed."""

    def __init__(self, client: KubeClient):
        self.client = client

    def _claim_name(self, ws: Workspace, i: int) -> str:
        return f"{ws.name}-az-{i}"

    def provision_nodes(self, ws, count):
        created = []
        for i in range(count):
            name = self._claim_name(ws, i)
            try:
                self.client.get("NodeClaim", ws.namespace, name)
            except NotFound:
                self.client.create({
                    "apiVersion": "karpenter.sh/v1",
                    "kind": "NodeClaim",
                    "metadata": {
                        "name": name, "namespace": ws.namespace,
                        "labels": {
                            LABEL_WORKSPACE_NAME: ws.name,
                            LABEL_WORKSPACE_NAMESPACE: ws.namespace,
                            "karpenter.sh/capacity-type": "on-demand",
                        },
                        "annotations": {
                            "kubernetes.azure.com/apiversion": "v1",
                        },
                    },
                    "spec": {
                        "nodeClassRef": {"group": "karpenter.azure.com",
                                         "kind": "AKSNodeClass",
                                         "name": "default"},
                        "requirements": [
                            {"key": "node.kubernetes.io/instance-type",
                             "operator": "In",
                             "values": [ws.resource.instanceType]},
                            {"key": "karpenter.sh/capacity-type",
                             "operator": "In", "values": ["on-demand"]},
                        ],
                        "resources": {"requests": {"amd.com/gpu": "1"}},
                    },
                    "status": {},
                })
                created.append(name)
        return created

    def ensure_nodes_ready(self, ws, count):
        ready = []
        for claim in self.client.list("NodeClaim", ws.namespace, {
                LABEL_WORKSPACE_NAME: ws.name}):
            node_name = claim.get("status", {}).get("nodeName")
            if not node_name:
                continue
            try:
                node = self.client.get("Node", "", node_name)
            except NotFound:
                continue
            if _node_ready(node):
                ready.append(node_name)
        return ready[:count]

    def delete_nodes(self, ws):
        for claim in self.client.list("NodeClaim", ws.namespace, {
                LABEL_WORKSPACE_NAME: ws.name}):
            self.client.delete("NodeClaim", ws.namespace,
                               claim["metadata"]["name"])


def make_provisioner(kind: str, client: KubeClient, **kw) -> NodeProvisioner:
    """Reference parity: nodeprovision/manager/factory.go:121."""
    if kind in ("byo", "none"):
        return BYOProvisioner(client)
    if kind == "karpenter":
        return KarpenterProvisioner(client, **kw)
    if kind in ("azure", "azure-gpu-provisioner", "gpuprovisioner"):
        return AzureGPUProvisioner(client)
    raise ValueError(f"unknown provisioner {kind!r}")
