"""InferenceSet reconciler — Python re-implementation of the reference's
pkg/inferenceset/inferenceset_controller.go (765 L): N replicas of a
Workspace template, upgrade-aware scale-down (old-revision-first, keep
Ready >= desired), label propagation, TPM aggregation, HPA/KEDA selector.
"""
from __future__ import annotations

import hashlib
import json
from dataclasses import dataclass
from typing import Dict, List

from ..api_types import (COND_INFERENCESET_READY, Condition, InferenceSet,
                         LABEL_INFERENCESET_CREATED_BY,
                         LABEL_UPGRADE_TO_VERSION)
from ..kubeclient import KubeClient


@dataclass
class ISReconcileResult:
    requeue_after_s: float = 0.0
    created: int = 0
    deleted: int = 0


def _ws_obj(iset: InferenceSet, index: int, revision: str) -> Dict:
    """Materialize the workspace template as a stored object dict."""
    tpl = iset.spec.workspaceTemplate
    return {
        "apiVersion": "kaito.sh/v1beta1",
        "kind": "Workspace",
        "metadata": {
            "name": f"{iset.name}-{index}",
            "namespace": iset.namespace,
            "labels": {
                LABEL_INFERENCESET_CREATED_BY: iset.name,
                "inferenceset.kaito.io/revision": revision,
            },
        },
        "spec": {
            "resource": {
                "instanceType": tpl.resource.instanceType,
                "count": tpl.resource.count,
            },
            "inference": {
                "preset": tpl.inference.preset.name
                if tpl.inference and tpl.inference.preset else None,
            },
        },
        "status": {},
    }


class InferenceSetReconciler:
    def __init__(self, client: KubeClient, gateway_api: bool = False):
        from ..expectations import ControllerExpectations
        self.expectations = ControllerExpectations()
        self.client = client
        self.gateway_api = gateway_api  # gatewayAPIInferenceExtension gate

    def _revision(self, iset: InferenceSet) -> str:
        tpl = iset.spec.workspaceTemplate
        payload = json.dumps({
            "instanceType": tpl.resource.instanceType,
            "preset": tpl.inference.preset.name
            if tpl.inference and tpl.inference.preset else None,
        }, sort_keys=True)
        return hashlib.sha256(payload.encode()).hexdigest()[:10]

    def _list_children(self, iset: InferenceSet) -> List[Dict]:
        return self.client.list("Workspace", iset.namespace, {
            LABEL_INFERENCESET_CREATED_BY: iset.name})

    @staticmethod
    def _is_ready(ws_obj: Dict) -> bool:
        return ws_obj.get("status", {}).get("state") == "Running"

    def _finalize(self, iset: InferenceSet) -> ISReconcileResult:
        """Deletion flow: remove child Workspaces, then the finalizer
        (reference: InferenceSetDeleting + finalizer drop)."""
        from ..api_types import (COND_INFERENCESET_DELETING,
                                 FINALIZER_INFERENCESET)
        res = ISReconcileResult()
        for obj in self._list_children(iset):
            self.client.delete("Workspace", iset.namespace,
                               obj["metadata"]["name"])
            res.deleted += 1
        found = False
        for c in iset.status.conditions:
            if c.type == COND_INFERENCESET_DELETING:
                c.status = "True"
                found = True
        if not found:
            iset.status.conditions.append(Condition(
                COND_INFERENCESET_DELETING, "True", "Deleting"))
        if FINALIZER_INFERENCESET in iset.finalizers:
            iset.finalizers.remove(FINALIZER_INFERENCESET)
        self.expectations.delete(f"{iset.namespace}/{iset.name}")
        return res

    def select_workspaces_to_delete(self, children: List[Dict], excess: int,
                                    revision: str) -> List[Dict]:
        """Reference parity: selectWorkspacesToDelete
        (inferenceset_controller.go:225-300) — delete old-revision first,
        then not-ready, then newest index; keep Ready >= desired."""
        def sort_key(obj):
            old_rev = obj["metadata"]["labels"].get(
                "inferenceset.kaito.io/revision") != revision
            ready = self._is_ready(obj)
            idx = int(obj["metadata"]["name"].rsplit("-", 1)[1])
            # delete priority: old revision first, then not-ready, then
            # highest index
            return (not old_rev, ready, -idx)

        return sorted(children, key=sort_key)[:excess]

    def reconcile(self, iset: InferenceSet) -> ISReconcileResult:
        if iset.deletionTimestamp:
            return self._finalize(iset)
        from ..api_types import FINALIZER_INFERENCESET
        if FINALIZER_INFERENCESET not in iset.finalizers:
            iset.finalizers.append(FINALIZER_INFERENCESET)
        iset.validate()
        revision = self._revision(iset)
        children = self._list_children(iset)
        desired = iset.spec.replicas
        res = ISReconcileResult()
        key = f"{iset.namespace}/{iset.name}"

        # cache-staleness guard (reference: ControllerExpectations,
        # inferenceset_controller.go:336-386): while a previous reconcile's
        # creates/deletes have not shown up in the store, do not scale
        # again — a stale child list would double-create or over-delete.
        if not self.expectations.satisfied(key):
            res.requeue_after_s = 1.0
        elif len(children) > desired:
            victims = self.select_workspaces_to_delete(
                children, len(children) - desired, revision)
            self.expectations.expect_deletions(key, len(victims))
            for obj in victims:
                self.client.delete("Workspace", iset.namespace,
                                   obj["metadata"]["name"])
                self.expectations.deletion_observed(key)
                res.deleted += 1
        elif len(children) < desired:
            used = {int(o["metadata"]["name"].rsplit("-", 1)[1])
                    for o in children}
            idx = 0
            self.expectations.expect_creations(key, desired - len(children))
            for _ in range(desired - len(children)):
                while idx in used:
                    idx += 1
                used.add(idx)
                self.client.create(_ws_obj(iset, idx, revision))
                self.expectations.creation_observed(key)
                res.created += 1

        # upgrade: children on an old revision get the upgrade label
        # (consumed by shouldUpgradeBaseImage, workspace_controller.go:695)
        for obj in self._list_children(iset):
            labels = obj["metadata"].setdefault("labels", {})
            if labels.get("inferenceset.kaito.io/revision") != revision and \
                    LABEL_UPGRADE_TO_VERSION not in labels:
                labels[LABEL_UPGRADE_TO_VERSION] = revision
                self.client.update(obj)

        if self.gateway_api:
            from ..manifests import (generate_inference_pool_helm_release,
                                     generate_inference_pool_oci_repository)
            self.client.apply(generate_inference_pool_oci_repository(
                iset.name, iset.namespace))
            self.client.apply(generate_inference_pool_helm_release(
                iset.name, iset.namespace))

        # status
        children = self._list_children(iset)
        ready = sum(1 for o in children if self._is_ready(o))
        iset.status.replicas = len(children)
        iset.status.readyReplicas = ready
        iset.status.selector = f"{LABEL_INFERENCESET_CREATED_BY}={iset.name}"
        # aggregated TPM (:177-193,429-514)
        tpm = 0.0
        for o in children:
            for m in o.get("status", {}).get("performance", {}) \
                    .get("metrics", []):
                if m.get("name") == "peakTokensPerMinute" and m.get("value"):
                    tpm += float(m["value"])
        iset.status.aggregatedPeakTokensPerMinute = tpm
        ok = ready >= desired
        found = False
        for c in iset.status.conditions:
            if c.type == COND_INFERENCESET_READY:
                c.status = "True" if ok else "False"
                found = True
        if not found:
            iset.status.conditions.append(Condition(
                COND_INFERENCESET_READY, "True" if ok else "False",
                "Ready" if ok else "ScalingOrWaiting"))
        if not ok:
            res.requeue_after_s = 5.0
        return res
