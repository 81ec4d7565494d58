"""ModelMirror reconciler — Python re-implementation of the reference's
pkg/modelmirror (modelmirror_controller.go:63-455): pre-download model
weights to a PVC so Workspace pods mount them instead of pulling.

Managed mode: ensure PVC → ensure download Job → track progress from pod
logs → phase Ready; Static mode: BYO weights path, immediately Ready.
Workspace gating (ensureModelMirror/waitForModelMirror,
workspace_controller.go:163-325) keys on status.phase == "Ready".
"""
from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Callable, Dict, Optional

from ..kubeclient import KubeClient, NotFound

PHASE_PENDING = "Pending"
PHASE_DOWNLOADING = "Downloading"
PHASE_READY = "Ready"
PHASE_FAILED = "Failed"


@dataclass
class ModelMirror:
    """api/v1alpha1/modelmirror_types.go:28-62 (cluster-scoped)."""
    name: str
    modelName: str = ""
    mode: str = "Managed"             # Managed | Static
    storageClassName: str = "kaito-local-nvme-disk"
    storageSize: str = "200Gi"
    staticVolumePath: str = ""
    namespace: str = "kaito-system"   # PVC/Job namespace
    status: Dict = field(default_factory=dict)


def classify_download_failure(logs: str) -> Optional[str]:
    """Reference parity: classifyDownloadFailure (:335)."""
    low = logs.lower()
    if "401" in low or "unauthorized" in low or "authentication" in low:
        return "AuthFailure"
    if "404" in low or "not found" in low:
        return "ModelNotFound"
    if "no space left" in low or "disk quota" in low:
        return "OutOfDisk"
    if "connection" in low or "timeout" in low:
        return "NetworkFailure"
    return "Unknown"


class ModelMirrorReconciler:
    def __init__(self, client: KubeClient, image: str = "ghcr.io/kaito-amd/downloader:latest",
                 get_pod_logs: Optional[Callable[[str, str], str]] = None):
        self.client = client
        self.image = image
        self.get_pod_logs = get_pod_logs or (lambda ns, name: "")

    def _pvc_name(self, mm: ModelMirror) -> str:
        return f"modelmirror-{mm.name}"

    def ensure_pvc(self, mm: ModelMirror):
        """Reference parity: ensurePVC (:168)."""
        name = self._pvc_name(mm)
        try:
            return self.client.get("PersistentVolumeClaim", mm.namespace, name)
        except NotFound:
            return self.client.create({
                "apiVersion": "v1", "kind": "PersistentVolumeClaim",
                "metadata": {"name": name, "namespace": mm.namespace,
                             "labels": {"kaito.sh/modelmirror": mm.name}},
                "spec": {
                    "accessModes": ["ReadWriteOnce"],
                    "storageClassName": mm.storageClassName,
                    "resources": {"requests": {"storage": mm.storageSize}},
                },
            })

    def ensure_download_job(self, mm: ModelMirror):
        """Reference parity: ensureDownloadJob (:217)."""
        name = f"modelmirror-{mm.name}-download"
        try:
            return self.client.get("Job", mm.namespace, name)
        except NotFound:
            return self.client.create({
                "apiVersion": "batch/v1", "kind": "Job",
                "metadata": {"name": name, "namespace": mm.namespace,
                             "labels": {"kaito.sh/modelmirror": mm.name}},
                "spec": {"backoffLimit": 3, "template": {"spec": {
                    "restartPolicy": "Never",
                    "containers": [{
                        "name": "downloader", "image": self.image,
                        "command": ["python3", "-m",
                                    "kaito_amd.utils.model_download",
                                    "--model", mm.modelName,
                                    "--dest", "/weights"],
                        "volumeMounts": [{"name": "weights",
                                          "mountPath": "/weights"}],
                    }],
                    "volumes": [{"name": "weights",
                                 "persistentVolumeClaim": {
                                     "claimName": self._pvc_name(mm)}}],
                }}},
                "status": {},
            })

    def reconcile(self, mm: ModelMirror) -> str:
        """Returns the phase written to status."""
        if mm.mode == "Static":
            mm.status = {"phase": PHASE_READY, "path": mm.staticVolumePath,
                         "progress": 100}
            return PHASE_READY
        self.ensure_pvc(mm)
        job = self.ensure_download_job(mm)
        st = job.get("status", {})
        if st.get("succeeded"):
            mm.status = {"phase": PHASE_READY,
                         "pvcName": self._pvc_name(mm), "progress": 100}
        elif st.get("failed", 0) > 3:
            logs = self.get_pod_logs(
                mm.namespace, f"modelmirror-{mm.name}-download")
            mm.status = {"phase": PHASE_FAILED,
                         "reason": classify_download_failure(logs)}
        elif st.get("active"):
            # progress sampling from pod logs (progress/progress.go)
            logs = self.get_pod_logs(
                mm.namespace, f"modelmirror-{mm.name}-download")
            pct = 0
            for m in re.finditer(r"(\d{1,3})%", logs):
                pct = max(pct, min(int(m.group(1)), 100))
            mm.status = {"phase": PHASE_DOWNLOADING, "progress": pct}
        else:
            mm.status = {"phase": PHASE_PENDING, "progress": 0}
        return mm.status["phase"]
