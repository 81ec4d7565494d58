"""In-cluster KubeClient backed by the `kubernetes` package.

The reconcilers talk to the KubeClient interface (kubeclient.py); tests
and the dev loop use FakeKubeClient. This wrapper is the in-cluster
implementation the reference gets from controller-runtime — a thin,
mechanical mapping onto the dynamic/CustomObjects API. The `kubernetes`
package is NOT present in the air-gapped build image, so this module is
import-gated and exercised only in a real cluster image; everything
above it is covered by the fake-client suites.

Kinds map to (group, version, plural):
  our CRDs under kaito.sh/v1beta1 (Workspace, InferenceSet, RAGEngine,
  MultiRoleInference, ModelMirror) and core/apps builtins the
  reconcilers emit (StatefulSet, Service, Job, Pod, Node, ConfigMap,
  PersistentVolumeClaim).
"""
from __future__ import annotations

from typing import Dict, List, Optional

from .kubeclient import KubeClient, NotFound, Obj

CRD_GROUP = "kaito.sh"
CRD_VERSION = "v1beta1"
CRD_PLURALS = {
    "Workspace": "workspaces",
    "InferenceSet": "inferencesets",
    "RAGEngine": "ragengines",
    "MultiRoleInference": "multiroleinferences",
    "ModelMirror": "modelmirrors",
    "OCIRepository": "ocirepositories",       # fluxcd source.toolkit
    "HelmRelease": "helmreleases",            # fluxcd helm.toolkit
    "NodeClaim": "nodeclaims",                # karpenter.sh
    "NodePool": "nodepools",
}
CRD_GROUPS = {
    "OCIRepository": ("source.toolkit.fluxcd.io", "v1"),
    "HelmRelease": ("helm.toolkit.fluxcd.io", "v2"),
    "NodeClaim": ("karpenter.sh", "v1"),
    "NodePool": ("karpenter.sh", "v1"),
}
BUILTIN_API = {
    # kind → (api attr on client module, namespaced list/get prefix)
    "StatefulSet": ("AppsV1Api", "stateful_set"),
    "Service": ("CoreV1Api", "service"),
    "Job": ("BatchV1Api", "job"),
    "Pod": ("CoreV1Api", "pod"),
    "ConfigMap": ("CoreV1Api", "config_map"),
    "PersistentVolumeClaim": ("CoreV1Api", "persistent_volume_claim"),
}


def _selector_str(sel: Optional[Dict[str, str]]) -> Optional[str]:
    if not sel:
        return None
    return ",".join(f"{k}={v}" for k, v in sorted(sel.items()))


class InClusterKubeClient(KubeClient):
    """Requires the `kubernetes` package and in-cluster (or kubeconfig)
    credentials."""

    def __init__(self):
        import kubernetes  # noqa: F401  (gated dependency)
        from kubernetes import client, config
        try:
            config.load_incluster_config()
        except Exception:  # noqa: BLE001 — dev fallback
            config.load_kube_config()
        self._client = client
        self._custom = client.CustomObjectsApi()

    # ---- kind routing ----
    def _crd_coords(self, kind: str):
        group, version = CRD_GROUPS.get(kind, (CRD_GROUP, CRD_VERSION))
        return group, version, CRD_PLURALS[kind]

    def _is_crd(self, kind: str) -> bool:
        return kind in CRD_PLURALS

    def _builtin(self, kind: str):
        api_name, prefix = BUILTIN_API[kind]
        return getattr(self._client, api_name)(), prefix

    # ---- interface ----
    def get(self, kind: str, namespace: str, name: str) -> Obj:
        from kubernetes.client.rest import ApiException
        try:
            if self._is_crd(kind):
                g, v, pl = self._crd_coords(kind)
                return self._custom.get_namespaced_custom_object(
                    g, v, namespace, pl, name)
            api, prefix = self._builtin(kind)
            fn = getattr(api, f"read_namespaced_{prefix}")
            return self._client.ApiClient().sanitize_for_serialization(
                fn(name, namespace))
        except ApiException as e:
            if e.status == 404:
                raise NotFound(f"{kind} {namespace}/{name}") from e
            raise

    def list(self, kind: str, namespace: Optional[str] = None,
             label_selector: Optional[Dict[str, str]] = None) -> List[Obj]:
        sel = _selector_str(label_selector)
        if self._is_crd(kind):
            g, v, pl = self._crd_coords(kind)
            if namespace:
                res = self._custom.list_namespaced_custom_object(
                    g, v, namespace, pl, label_selector=sel)
            else:
                res = self._custom.list_cluster_custom_object(
                    g, v, pl, label_selector=sel)
            return res.get("items", [])
        api, prefix = self._builtin(kind)
        if namespace:
            fn = getattr(api, f"list_namespaced_{prefix}")
            res = fn(namespace, label_selector=sel)
        else:
            fn = getattr(api, f"list_{prefix}_for_all_namespaces")
            res = fn(label_selector=sel)
        san = self._client.ApiClient().sanitize_for_serialization
        return [san(it) for it in res.items]

    def create(self, obj: Obj) -> Obj:
        kind = obj["kind"]
        ns = obj["metadata"].get("namespace", "default")
        if self._is_crd(kind):
            g, v, pl = self._crd_coords(kind)
            return self._custom.create_namespaced_custom_object(
                g, v, ns, pl, obj)
        api, prefix = self._builtin(kind)
        fn = getattr(api, f"create_namespaced_{prefix}")
        return self._client.ApiClient().sanitize_for_serialization(
            fn(ns, obj))

    def update(self, obj: Obj) -> Obj:
        kind = obj["kind"]
        md = obj["metadata"]
        ns = md.get("namespace", "default")
        if self._is_crd(kind):
            g, v, pl = self._crd_coords(kind)
            return self._custom.replace_namespaced_custom_object(
                g, v, ns, pl, md["name"], obj)
        api, prefix = self._builtin(kind)
        fn = getattr(api, f"replace_namespaced_{prefix}")
        return self._client.ApiClient().sanitize_for_serialization(
            fn(md["name"], ns, obj))

    def update_status(self, obj: Obj) -> Obj:
        kind = obj["kind"]
        md = obj["metadata"]
        ns = md.get("namespace", "default")
        if self._is_crd(kind):
            g, v, pl = self._crd_coords(kind)
            return self._custom.replace_namespaced_custom_object_status(
                g, v, ns, pl, md["name"], obj)
        api, prefix = self._builtin(kind)
        fn = getattr(api, f"replace_namespaced_{prefix}_status", None)
        if fn is None:
            return self.update(obj)
        return self._client.ApiClient().sanitize_for_serialization(
            fn(md["name"], ns, obj))

    def delete(self, kind: str, namespace: str, name: str) -> None:
        from kubernetes.client.rest import ApiException
        try:
            if self._is_crd(kind):
                g, v, pl = self._crd_coords(kind)
                self._custom.delete_namespaced_custom_object(
                    g, v, namespace, pl, name)
                return
            api, prefix = self._builtin(kind)
            getattr(api, f"delete_namespaced_{prefix}")(name, namespace)
        except ApiException as e:
            if e.status != 404:
                raise


def make_kube_client():
    """In-cluster client when `kubernetes` is importable, else the
    in-memory fake (dev/test)."""
    try:
        return InClusterKubeClient()
    except ImportError:
        from .kubeclient import FakeKubeClient
        return FakeKubeClient()
