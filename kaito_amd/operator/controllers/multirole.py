"""MultiRoleInference reconciler — Python re-implementation of the
reference's pkg/controllers/multiroleinference/controller.go (784 L):
prefill/decode disaggregation as one child InferenceSet per role with
kaito.sh/inference-role labels, shared routing (InferencePool + EPP with
prefill-filter/decode-filter plugins), aggregated status.

Engine-side counterpart: decode pods serve on :5001 behind a routing
sidecar on :5000; KV blocks move prefill→decode over the transfer channel
(kaito_amd.engine.kv_transfer — xGMI p2p same-node, TCP cross-node).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict

from ..api_types import (LABEL_INFERENCE_ROLE, LABEL_MRI_CREATED_BY,
                         ValidationError)
from ..kubeclient import KubeClient, NotFound

ROLE_PREFILL = "prefill"
ROLE_DECODE = "decode"


@dataclass
class RoleSpec:
    replicas: int = 1
    instanceType: str = ""


@dataclass
class MultiRoleInference:
    """api/v1alpha1/multiroleinference_types.go."""
    name: str
    namespace: str = "default"
    preset: str = ""
    prefill: RoleSpec = field(default_factory=RoleSpec)
    decode: RoleSpec = field(default_factory=RoleSpec)
    status: Dict = field(default_factory=dict)

    def validate(self):
        if not self.preset:
            raise ValidationError("preset required")
        if self.prefill.replicas < 1 or self.decode.replicas < 1:
            raise ValidationError("both roles need >=1 replica")


class MultiRoleInferenceReconciler:
    def __init__(self, client: KubeClient):
        self.client = client

    def _child_name(self, mri: MultiRoleInference, role: str) -> str:
        return f"{mri.name}-{role}"

    def _ensure_role_set(self, mri: MultiRoleInference, role: str,
                         spec: RoleSpec):
        name = self._child_name(mri, role)
        obj = {
            "apiVersion": "kaito.sh/v1beta1", "kind": "InferenceSet",
            "metadata": {
                "name": name, "namespace": mri.namespace,
                "labels": {LABEL_MRI_CREATED_BY: mri.name,
                           LABEL_INFERENCE_ROLE: role},
            },
            "spec": {
                "replicas": spec.replicas,
                "workspaceTemplate": {
                    "resource": {"instanceType": spec.instanceType},
                    "inference": {"preset": mri.preset},
                    "labels": {LABEL_INFERENCE_ROLE: role},
                    # decode pods: engine on 5001, routing sidecar on 5000
                    # (reference: consts.go:173-183)
                    "env": {"KAITO_INFERENCE_ROLE": role},
                },
            },
            "status": {},
        }
        try:
            existing = self.client.get("InferenceSet", mri.namespace, name)
            if existing["spec"]["replicas"] != spec.replicas:
                existing["spec"]["replicas"] = spec.replicas
                self.client.update(existing)
        except NotFound:
            self.client.create(obj)

    def _ensure_routing(self, mri: MultiRoleInference):
        """Shared InferencePool + EPP config with P/D plugins
        (reference :560-575)."""
        name = f"{mri.name}-pool"
        pool = {
            "apiVersion": "inference.networking.x-k8s.io/v1alpha2",
            "kind": "InferencePool",
            "metadata": {"name": name, "namespace": mri.namespace,
                         "labels": {LABEL_MRI_CREATED_BY: mri.name}},
            "spec": {
                "selector": {LABEL_MRI_CREATED_BY: mri.name},
                "targetPort": 5000,
                "eppConfig": {
                    "plugins": [
                        {"name": "prefill-filter",
                         "match": {LABEL_INFERENCE_ROLE: ROLE_PREFILL}},
                        {"name": "decode-filter",
                         "match": {LABEL_INFERENCE_ROLE: ROLE_DECODE}},
                        {"name": "load-aware-scorer"},
                        {"name": "kv-cache-utilization-scorer"},
                    ],
                },
            },
        }
        self.client.apply(pool)

    def _aggregate_status(self, mri: MultiRoleInference) -> Dict:
        """Reference parity: aggregateStatus (:255)."""
        out = {}
        ready_total = want_total = 0
        for role in (ROLE_PREFILL, ROLE_DECODE):
            try:
                child = self.client.get("InferenceSet", mri.namespace,
                                        self._child_name(mri, role))
            except NotFound:
                continue
            st = child.get("status", {})
            ready = st.get("readyReplicas", 0)
            want = child["spec"]["replicas"]
            out[role] = {"readyReplicas": ready, "replicas": want}
            ready_total += ready
            want_total += want
        out["ready"] = ready_total >= want_total and want_total > 0
        return out

    def reconcile(self, mri: MultiRoleInference) -> Dict:
        mri.validate()
        self._ensure_role_set(mri, ROLE_PREFILL, mri.prefill)
        self._ensure_role_set(mri, ROLE_DECODE, mri.decode)
        self._ensure_routing(mri)
        mri.status = self._aggregate_status(mri)
        return mri.status
