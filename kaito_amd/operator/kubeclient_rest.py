"""Dependency-free in-cluster KubeClient speaking the Kubernetes REST
API directly over httpx.

The reference links controller-runtime; the first in-cluster wrapper
here (kubeclient_incluster.py) maps onto the `kubernetes` package —
which is NOT in the air-gapped serving image. This client removes that
dependency: service-account bearer auth + CA bundle from the standard
projected paths, and the plain REST scheme

  CRDs:      /apis/{group}/{version}/namespaces/{ns}/{plural}[/{name}]
  core:      /api/v1/namespaces/{ns}/{plural}[/{name}]
  apps/batch:/apis/{group}/v1/namespaces/{ns}/{plural}[/{name}]
  status:    PATCH {object}/status (merge-patch)

Validated against a mock API server in tests/test_kubeclient_rest.py
(the closest an air-gapped environment gets to envtest).

Env (all overridable for tests):
  KUBERNETES_SERVICE_HOST / KUBERNETES_SERVICE_PORT — in-cluster target
  KUBE_API_URL          — full base URL override (tests / kubeconfigless)
  KUBE_TOKEN_FILE       — bearer token path (default: the SA projection)
  KUBE_CA_FILE          — CA bundle path ("" disables verification)
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Optional, Tuple

from .kubeclient import Conflict, KubeClient, NotFound, Obj
from .kubeclient_incluster import (CRD_GROUP, CRD_GROUPS, CRD_PLURALS,
                                   CRD_VERSION, _selector_str)

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"

# builtin kind → (api prefix, plural)
BUILTIN_PATHS: Dict[str, Tuple[str, str]] = {
    "Pod": ("/api/v1", "pods"),
    "Service": ("/api/v1", "services"),
    "ConfigMap": ("/api/v1", "configmaps"),
    "Secret": ("/api/v1", "secrets"),
    "PersistentVolumeClaim": ("/api/v1", "persistentvolumeclaims"),
    "Node": ("/api/v1", "nodes"),
    "StatefulSet": ("/apis/apps/v1", "statefulsets"),
    "Deployment": ("/apis/apps/v1", "deployments"),
    "Job": ("/apis/batch/v1", "jobs"),
    "Lease": ("/apis/coordination.k8s.io/v1", "leases"),
}
CLUSTER_SCOPED = {"Node", "NodeClaim", "NodePool"}


class InClusterRestClient(KubeClient):
    def __init__(self, base_url: Optional[str] = None):
        import httpx
        url = base_url or os.environ.get("KUBE_API_URL")
        if not url:
            host = os.environ.get("KUBERNETES_SERVICE_HOST")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            if not host:
                raise RuntimeError("not running in-cluster "
                                   "(KUBERNETES_SERVICE_HOST unset)")
            url = f"https://{host}:{port}"
        headers = {"Content-Type": "application/json"}
        token_file = os.environ.get("KUBE_TOKEN_FILE",
                                    os.path.join(SA_DIR, "token"))
        if os.path.exists(token_file):
            with open(token_file) as f:
                headers["Authorization"] = f"Bearer {f.read().strip()}"
        ca = os.environ.get("KUBE_CA_FILE", os.path.join(SA_DIR, "ca.crt"))
        verify = ca if ca and os.path.exists(ca) else False
        self._http = httpx.Client(base_url=url, headers=headers,
                                  verify=verify, timeout=30)

    # ---------------------------------------------------------- paths
    def _path(self, kind: str, namespace: Optional[str],
              name: Optional[str] = None) -> str:
        if kind in CRD_PLURALS:
            group, version = CRD_GROUPS.get(kind, (CRD_GROUP, CRD_VERSION))
            plural = CRD_PLURALS[kind]
            prefix = f"/apis/{group}/{version}"
        elif kind in BUILTIN_PATHS:
            prefix, plural = BUILTIN_PATHS[kind]
        else:
            raise ValueError(f"unmapped kind {kind}")
        if kind in CLUSTER_SCOPED or not namespace:
            p = f"{prefix}/{plural}"
        else:
            p = f"{prefix}/namespaces/{namespace}/{plural}"
        return f"{p}/{name}" if name else p

    def _check(self, r, kind: str, name: str):
        if r.status_code == 404:
            raise NotFound(f"{kind}/{name}")
        if r.status_code == 409:
            raise Conflict(f"{kind}/{name}")
        if r.status_code >= 400:
            raise RuntimeError(
                f"kube API {r.request.method} {r.request.url.path}: "
                f"{r.status_code} {r.text[:200]}")
        return r

    # ------------------------------------------------------- interface
    def get(self, kind: str, namespace: str, name: str) -> Obj:
        r = self._http.get(self._path(kind, namespace, name))
        return self._check(r, kind, name).json()

    def list(self, kind: str, namespace: Optional[str] = None,
             label_selector: Optional[Dict[str, str]] = None) -> List[Obj]:
        params = {}
        sel = _selector_str(label_selector)
        if sel:
            params["labelSelector"] = sel
        r = self._http.get(self._path(kind, namespace), params=params)
        return self._check(r, kind, "<list>").json().get("items", [])

    def create(self, obj: Obj) -> Obj:
        md = obj.get("metadata", {})
        r = self._http.post(self._path(obj["kind"], md.get("namespace")),
                            content=json.dumps(obj))
        return self._check(r, obj["kind"], md.get("name", "?")).json()

    def update(self, obj: Obj) -> Obj:
        md = obj.get("metadata", {})
        r = self._http.put(
            self._path(obj["kind"], md.get("namespace"), md.get("name")),
            content=json.dumps(obj))
        return self._check(r, obj["kind"], md.get("name", "?")).json()

    def update_status(self, obj: Obj) -> Obj:
        md = obj.get("metadata", {})
        path = self._path(obj["kind"], md.get("namespace"),
                          md.get("name")) + "/status"
        r = self._http.patch(
            path, content=json.dumps({"status": obj.get("status", {})}),
            headers={"Content-Type": "application/merge-patch+json"})
        return self._check(r, obj["kind"], md.get("name", "?")).json()

    def delete(self, kind: str, namespace: str, name: str) -> None:
        r = self._http.delete(self._path(kind, namespace, name))
        if r.status_code != 404:
            self._check(r, kind, name)


def make_kube_client() -> KubeClient:
    """Prefer the dependency-free REST client in-cluster; fall back to
    the `kubernetes`-package wrapper if that import path is forced."""
    return InClusterRestClient()
