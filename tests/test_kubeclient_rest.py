"""InClusterRestClient vs a mock Kubernetes API server.

Spins a real HTTP server implementing the k8s REST scheme (namespaced
CRD + builtin paths, labelSelector lists, /status merge-patch, 404/409
semantics) and drives the dependency-free client — then runs a REAL
reconciler (workspace controller) against it end-to-end. The air-gapped
analog of envtest (reference: pkg/utils/test/mock_client.go strategy).
"""
import http.server
import json
import threading
from urllib.parse import parse_qs, urlparse

import pytest

from kaito_amd.operator.kubeclient import Conflict, NotFound
from kaito_amd.operator.kubeclient_rest import InClusterRestClient


class _MockKubeAPI(http.server.BaseHTTPRequestHandler):
    store = {}     # path -> obj
    rv = [0]

    # -------------------------------------------------- helpers
    def _send(self, code, obj):
        out = json.dumps(obj).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(out)))
        self.end_headers()
        self.wfile.write(out)

    def _body(self):
        n = int(self.headers.get("Content-Length", 0))
        return json.loads(self.rfile.read(n)) if n else {}

    def log_message(self, *a):
        pass

    # -------------------------------------------------- verbs
    PLURALS = {"workspaces", "inferencesets", "ragengines",
               "multiroleinferences", "modelmirrors", "pods", "services",
               "configmaps", "secrets", "persistentvolumeclaims", "nodes",
               "statefulsets", "deployments", "jobs", "leases",
               "nodeclaims", "nodepools", "ocirepositories",
               "helmreleases"}

    def do_GET(self):
        u = urlparse(self.path)
        if u.path in self.store:
            return self._send(200, self.store[u.path])
        if u.path.rstrip("/").rsplit("/", 1)[-1] not in self.PLURALS:
            return self._send(404, {"reason": "NotFound"})   # object GET
        # collection list
        prefix = u.path.rstrip("/") + "/"
        items = [o for p, o in self.store.items() if
                 p.startswith(prefix) and "/" not in p[len(prefix):]]
        sel = parse_qs(u.query).get("labelSelector", [None])[0]
        if sel:
            want = dict(kv.split("=", 1) for kv in sel.split(","))
            items = [o for o in items
                     if all(o.get("metadata", {}).get("labels", {})
                            .get(k) == v for k, v in want.items())]
        self._send(200, {"kind": "List", "items": items})

    def do_POST(self):
        obj = self._body()
        name = obj["metadata"]["name"]
        path = urlparse(self.path).path.rstrip("/") + "/" + name
        if path in self.store:
            return self._send(409, {"reason": "AlreadyExists"})
        self.rv[0] += 1
        obj["metadata"]["resourceVersion"] = str(self.rv[0])
        self.store[path] = obj
        self._send(201, obj)

    def do_PUT(self):
        path = urlparse(self.path).path
        if path not in self.store:
            return self._send(404, {"reason": "NotFound"})
        obj = self._body()
        self.rv[0] += 1
        obj["metadata"]["resourceVersion"] = str(self.rv[0])
        self.store[path] = obj
        self._send(200, obj)

    def do_PATCH(self):
        path = urlparse(self.path).path
        if not path.endswith("/status"):
            return self._send(404, {"reason": "NotFound"})
        objpath = path[: -len("/status")]
        if objpath not in self.store:
            return self._send(404, {"reason": "NotFound"})
        patch = self._body()
        obj = self.store[objpath]
        obj["status"] = patch.get("status", {})
        self.rv[0] += 1
        obj["metadata"]["resourceVersion"] = str(self.rv[0])
        self._send(200, obj)

    def do_DELETE(self):
        path = urlparse(self.path).path
        if path not in self.store:
            return self._send(404, {"reason": "NotFound"})
        del self.store[path]
        self._send(200, {"status": "Success"})


@pytest.fixture()
def kube(monkeypatch):
    _MockKubeAPI.store = {}
    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), _MockKubeAPI)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    monkeypatch.setenv("KUBE_API_URL",
                       f"http://127.0.0.1:{srv.server_address[1]}")
    monkeypatch.setenv("KUBE_TOKEN_FILE", "/nonexistent")
    monkeypatch.setenv("KUBE_CA_FILE", "")
    try:
        yield InClusterRestClient()
    finally:
        srv.shutdown()


def test_crud_roundtrip_crd(kube):
    ws = {"apiVersion": "kaito.sh/v1beta1", "kind": "Workspace",
          "metadata": {"name": "ws1", "namespace": "default",
                       "labels": {"app": "kaito"}},
          "spec": {"instanceType": "Standard_ND96isr_MI355X_v6"}}
    created = kube.create(ws)
    assert created["metadata"]["resourceVersion"] == "1"
    with pytest.raises(Conflict):
        kube.create(ws)

    got = kube.get("Workspace", "default", "ws1")
    assert got["spec"]["instanceType"].endswith("MI355X_v6")

    got["spec"]["instanceType"] = "byo"
    kube.update(got)
    assert kube.get("Workspace", "default", "ws1")["spec"][
        "instanceType"] == "byo"

    got["status"] = {"phase": "Ready"}
    kube.update_status(got)
    refreshed = kube.get("Workspace", "default", "ws1")
    assert refreshed["status"]["phase"] == "Ready"
    assert refreshed["spec"]["instanceType"] == "byo"  # spec untouched

    kube.delete("Workspace", "default", "ws1")
    with pytest.raises(NotFound):
        kube.get("Workspace", "default", "ws1")
    kube.delete("Workspace", "default", "ws1")  # idempotent


def test_list_label_selector_and_builtin_paths(kube):
    for i, lbl in enumerate(["a", "a", "b"]):
        kube.create({"apiVersion": "v1", "kind": "Pod",
                     "metadata": {"name": f"p{i}", "namespace": "ns1",
                                  "labels": {"grp": lbl}},
                     "spec": {}})
    assert len(kube.list("Pod", "ns1")) == 3
    assert len(kube.list("Pod", "ns1", {"grp": "a"})) == 2
    # builtin path shape
    assert "/api/v1/namespaces/ns1/pods" == kube._path("Pod", "ns1")
    assert "/apis/apps/v1/namespaces/x/statefulsets/s" == \
        kube._path("StatefulSet", "x", "s")
    assert "/apis/kaito.sh/v1beta1/namespaces/x/workspaces" == \
        kube._path("Workspace", "x")
    assert kube._path("NodeClaim", None).startswith("/apis/karpenter.sh/")


def test_workspace_reconciler_against_mock_api(kube):
    """A REAL reconciler pass over the REST client + mock API server:
    the Workspace controller's status/finalizer writes must land on the
    (mock) API server through the wire — the air-gapped envtest
    analog upgrading kubeclient coverage beyond the in-memory fake."""
    from kaito_amd.operator import api_types as at
    from kaito_amd.operator.controllers.workspace import WorkspaceReconciler
    from kaito_amd.operator.nodeprovision import BYOProvisioner
    from kaito_amd.operator.sku import get_sku_handler
    kube.create({"apiVersion": "kaito.sh/v1beta1", "kind": "Workspace",
                 "metadata": {"name": "ws1", "namespace": "default"},
                 "spec": {}, "status": {}})
    ws = at.Workspace(
        name="ws1",
        resource=at.ResourceSpec(
            instanceType="Standard_ND96isr_MI355X_v6", count=1),
        inference=at.InferenceSpec(preset=at.PresetSpec(
            name="phi-4-mini-instruct")))
    rec = WorkspaceReconciler(kube, get_sku_handler("azure"),
                              BYOProvisioner(kube))
    res = rec.reconcile(ws)
    assert res.requeue   # BYO: waiting for matching ready nodes
    got = kube.get("Workspace", "default", "ws1")
    conds = {c["type"]: c["status"]
             for c in got["status"]["conditions"]}
    assert conds.get("NodesReady") == "False"
    assert got["status"]["state"] == "Pending"
    assert got["metadata"].get("finalizers"), "finalizer not persisted"
