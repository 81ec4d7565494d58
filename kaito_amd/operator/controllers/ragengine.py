"""RAGEngine reconciler — Python re-implementation of the reference's
pkg/ragengine/controllers (ragengine_controller.go:82 Reconcile,
preset_rag.go:198 CreatePresetRAG): RAGEngine CR → Deployment + Service
with the CRD spec wired into the service's env contract
(pkg/ragengine/manifests/manifests.go:155; env names in SURVEY.md §8),
guardrails policy ConfigMap, storage volume, GPU node path for local
embedding, and ServiceReady/RAGEngineSucceeded conditions.
"""
from __future__ import annotations

import time
from typing import Dict, List, Optional

from ..api_types import COND_RAGENGINE_SERVICE_READY, RAGEngine
from ..kubeclient import KubeClient, NotFound

RAG_PORT = 5000
LABEL_RAGENGINE = "kaito.sh/ragengine"


def rag_env_from_spec(rag: RAGEngine) -> List[Dict[str, str]]:
    """CRD spec → env contract (byte-compatible names, config.py)."""
    spec = rag.spec
    emb = spec.embedding or {}
    env: List[Dict[str, str]] = []

    def add(name, value):
        if value is not None and value != "":
            env.append({"name": name, "value": str(value)})

    if emb.get("local"):
        add("EMBEDDING_SOURCE_TYPE", "local")
        add("LOCAL_EMBEDDING_MODEL_ID",
            emb["local"].get("modelID", "BAAI/bge-small-en-v1.5"))
    elif emb.get("remote"):
        add("EMBEDDING_SOURCE_TYPE", "remote")
        add("REMOTE_EMBEDDING_URL", emb["remote"].get("url"))
        add("REMOTE_EMBEDDING_ACCESS_SECRET",
            emb["remote"].get("accessSecret"))
    st = spec.storage or {}
    add("VECTOR_DB_TYPE", st.get("vectorDBType", "faiss"))
    add("VECTOR_DB_URL", st.get("vectorDBURL"))
    add("DEFAULT_VECTOR_DB_PERSIST_DIR",
        st.get("persistDir", "/data/persist"))
    inf = spec.inferenceService or {}
    add("LLM_INFERENCE_URL", inf.get("url"))
    add("LLM_ACCESS_SECRET", inf.get("accessSecret"))
    add("LLM_CONTEXT_WINDOW", inf.get("contextWindow", 8192))
    gr = getattr(spec, "guardrails", None) or {}
    if gr.get("enabled"):
        add("OUTPUT_GUARDRAILS_ENABLED", "true")
        add("OUTPUT_GUARDRAILS_POLICY_PATH", "/etc/guardrails/policy.yaml")
        add("OUTPUT_GUARDRAILS_HOT_RELOAD_ENABLED",
            "true" if gr.get("hotReload") else "false")
    return env


class RAGEngineReconciler:
    def __init__(self, client: KubeClient,
                 image: str = "ghcr.io/kaito-amd/ragengine:latest"):
        self.client = client
        self.image = image

    def _deployment(self, rag: RAGEngine) -> Dict:
        sel = {LABEL_RAGENGINE: rag.name}
        local_embed = bool((rag.spec.embedding or {}).get("local"))
        container: Dict = {
            "name": rag.name,
            "image": self.image,
            "command": ["python3", "-m", "kaito_amd.ragengine.service"],
            "env": rag_env_from_spec(rag),
            "ports": [{"containerPort": RAG_PORT, "name": "http"}],
            "readinessProbe": {"httpGet": {"path": "/health",
                                           "port": RAG_PORT},
                               "periodSeconds": 10},
            "volumeMounts": [{"name": "storage", "mountPath": "/data"}],
        }
        if local_embed:
            # local embedding runs on a GPU node (ragengine_controller.go:368)
            container["resources"] = {"requests": {"amd.com/gpu": "1"},
                                      "limits": {"amd.com/gpu": "1"}}
        volumes: List[Dict] = []
        st = rag.spec.storage or {}
        if st.get("pvcName"):
            volumes.append({"name": "storage", "persistentVolumeClaim":
                            {"claimName": st["pvcName"]}})
        else:
            volumes.append({"name": "storage", "emptyDir": {}})
        gr = getattr(rag.spec, "guardrails", None) or {}
        if gr.get("enabled"):
            volumes.append({"name": "guardrails-policy", "configMap":
                            {"name": f"{rag.name}-guardrails"}})
            container["volumeMounts"].append(
                {"name": "guardrails-policy", "mountPath": "/etc/guardrails"})
        spec: Dict = {
            "replicas": 1,
            "selector": {"matchLabels": sel},
            "template": {"metadata": {"labels": sel}, "spec": {
                "containers": [container], "volumes": volumes}},
        }
        if rag.spec.compute and rag.spec.compute.instanceType and local_embed:
            spec["template"]["spec"]["nodeSelector"] = {
                "node.kubernetes.io/instance-type":
                    rag.spec.compute.instanceType}
        return {"apiVersion": "apps/v1", "kind": "Deployment",
                "metadata": {"name": rag.name, "namespace": rag.namespace,
                             "labels": sel},
                "spec": spec}

    def _service(self, rag: RAGEngine) -> Dict:
        return {"apiVersion": "v1", "kind": "Service",
                "metadata": {"name": rag.name, "namespace": rag.namespace,
                             "labels": {LABEL_RAGENGINE: rag.name}},
                "spec": {"type": "ClusterIP",
                         "selector": {LABEL_RAGENGINE: rag.name},
                         "ports": [{"port": 80, "targetPort": RAG_PORT}]}}

    def _guardrails_configmap(self, rag: RAGEngine) -> Optional[Dict]:
        gr = getattr(rag.spec, "guardrails", None) or {}
        if not gr.get("enabled"):
            return None
        return {"apiVersion": "v1", "kind": "ConfigMap",
                "metadata": {"name": f"{rag.name}-guardrails",
                             "namespace": rag.namespace},
                "data": {"policy.yaml": gr.get("policy", "")}}

    def reconcile(self, rag: RAGEngine) -> bool:
        """Returns True when the service is ready."""
        if getattr(rag, "deletionTimestamp", None):
            # deletion flow: tear down Deployment/Service/ConfigMap and
            # mark RAGEngineDeleting (reference condition_types.go)
            from ..api_types import COND_RAGENGINE_DELETING
            for kind in ("Deployment", "Service", "ConfigMap"):
                try:
                    self.client.delete(kind, rag.namespace, rag.name)
                except NotFound:
                    pass
            conds = rag.status.setdefault("conditions", [])
            if not any(c["type"] == COND_RAGENGINE_DELETING for c in conds):
                conds.append({"type": COND_RAGENGINE_DELETING,
                              "status": "True", "reason": "Deleting"})
            return False
        rag.validate()
        cm = self._guardrails_configmap(rag)
        if cm is not None:
            self.client.apply(cm)
        self.client.apply(self._deployment(rag))
        self.client.apply(self._service(rag))
        try:
            live = self.client.get("Deployment", rag.namespace, rag.name)
        except NotFound:
            live = {}
        ready = live.get("status", {}).get("readyReplicas", 0) >= 1
        conds = rag.status.setdefault("conditions", [])
        now = time.strftime("%Y-%m-%dT%H:%M:%SZ")
        for c in conds:
            if c["type"] == COND_RAGENGINE_SERVICE_READY:
                c.update(status="True" if ready else "False",
                         lastTransitionTime=now)
                break
        else:
            conds.append({"type": COND_RAGENGINE_SERVICE_READY,
                          "status": "True" if ready else "False",
                          "lastTransitionTime": now})
        rag.status["state"] = "Ready" if ready else "Pending"
        return ready
