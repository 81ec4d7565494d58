"""Controller reconcile tests against FakeKubeClient — the reference's
MockClient unit-test strategy: no cluster, full reconcile
paths, golden assertions on created objects and conditions."""
import json

import pytest

from kaito_amd.operator import api_types as at
from kaito_amd.operator.controllers.inferenceset import InferenceSetReconciler
from kaito_amd.operator.controllers.workspace import (WorkspaceReconciler,
                                                      classify_pod_failure)
from kaito_amd.operator.kubeclient import FakeKubeClient, NotFound
from kaito_amd.operator.nodeprovision import (BYOProvisioner,
                                              KarpenterProvisioner,
                                              make_provisioner)
from kaito_amd.operator.sku import get_sku_handler

SKU = "Standard_ND96isr_MI355X_v1"


def _node(name, ready=True, labels=None):
    return {"apiVersion": "v1", "kind": "Node",
            "metadata": {"name": name, "namespace": "", "labels": labels or {}},
            "status": {"conditions": [
                {"type": "Ready", "status": "True" if ready else "False"}]}}


def _ws(name="ws1", preset="llama-3-8b", count=1):
    return at.Workspace(
        name=name,
        resource=at.ResourceSpec(instanceType=SKU, count=count),
        inference=at.InferenceSpec(preset=at.PresetSpec(name=preset)))


def _store_ws(client, ws):
    client.create({"apiVersion": "kaito.sh/v1beta1", "kind": "Workspace",
                   "metadata": {"name": ws.name, "namespace": ws.namespace},
                   "spec": {}, "status": {}})


@pytest.fixture
def client():
    return FakeKubeClient()


def _reconciler(client, **kw):
    prov = BYOProvisioner(client)
    return WorkspaceReconciler(client, get_sku_handler("azure"), prov, **kw)


# ------------------------------------------------------------ workspace
def test_workspace_waits_for_nodes(client):
    ws = _ws()
    _store_ws(client, ws)
    r = _reconciler(client)
    res = r.reconcile(ws)
    assert res.requeue and res.requeue_after_s == 2.0
    assert ws.status.state == "Pending"
    conds = {c.type: c.status for c in ws.status.conditions}
    assert conds["NodesReady"] == "False"


def test_workspace_full_reconcile_to_running(client):
    ws = _ws()
    _store_ws(client, ws)
    client.create(_node("n1", labels={
        "node.kubernetes.io/instance-type": SKU}))
    r = _reconciler(client)
    res = r.reconcile(ws)  # creates statefulset; pods not ready yet
    assert not res.done
    ss = client.get("StatefulSet", "default", "ws1")
    assert ss["spec"]["replicas"] == 1
    assert client.get("Service", "default", "ws1")
    assert client.get("Service", "default", "ws1-headless")
    # simulate statefulset becoming ready
    ss["status"] = {"readyReplicas": 1}
    client.update(ss)
    res = r.reconcile(ws)
    assert res.done
    assert ws.status.state == "Running"
    assert ws.status.workerNodes == ["n1"]
    conds = {c.type: c.status for c in ws.status.conditions}
    assert conds["InferenceReady"] == "True"
    # status pushed to the stored object
    stored = client.get("Workspace", "default", "ws1")
    assert stored["status"]["state"] == "Running"


def test_workspace_estimator_sets_target_node_count(client):
    ws = _ws(preset="llama-3-70b")
    _store_ws(client, ws)
    r = _reconciler(client)
    r.reconcile(ws)
    assert ws.status.targetNodeCount == 1  # 70B fits one MI355X node


def test_workspace_pod_failure_classification(client):
    ws = _ws()
    _store_ws(client, ws)
    client.create(_node("n1", labels={
        "node.kubernetes.io/instance-type": SKU}))
    client.create({
        "apiVersion": "v1", "kind": "Pod",
        "metadata": {"name": "ws1-0", "namespace": "default",
                     "labels": {at.LABEL_WORKSPACE_NAME: "ws1"}},
        "status": {"phase": "Running", "containerStatuses": [{
            "state": {"waiting": {"reason": "CrashLoopBackOff"}},
            "lastState": {"terminated": {"reason": "OOMKilled"}}}]},
    })
    r = _reconciler(client)
    res = r.reconcile(ws)
    assert ws.status.state == "Failed"
    conds = {c.type: c.reason for c in ws.status.conditions}
    assert conds["InferenceReady"] == "OOMKilled"


def test_classify_variants():
    assert classify_pod_failure({"status": {"phase": "Pending", "conditions": [
        {"type": "PodScheduled", "status": "False"}]}}) == "Unschedulable"
    assert classify_pod_failure({"status": {"containerStatuses": [
        {"state": {"waiting": {"reason": "ImagePullBackOff"}}}]}}) == \
        "ImagePullFailure"
    assert classify_pod_failure({"status": {"phase": "Running"}}) is None


def test_workspace_benchmark_ingestion(client):
    ws = _ws()
    _store_ws(client, ws)
    client.create(_node("n1", labels={
        "node.kubernetes.io/instance-type": SKU}))
    logs = ("startup...\n"
            'KAITO_BENCHMARK_CONFIG: {"engine": "kaito-amd", "engineVersion": "0.1"}\n'
            'KAITO_BENCHMARK_RESULT: {"peakTokensPerMinute": 1523400}\n')
    r = _reconciler(client, get_pod_logs=lambda ns, name: logs)
    r.reconcile(ws)
    ss = client.get("StatefulSet", "default", "ws1")
    ss["status"] = {"readyReplicas": 1}
    client.update(ss)
    r.reconcile(ws)
    metrics = ws.status.performance["metrics"]
    assert metrics[0]["name"] == "peakTokensPerMinute"
    assert metrics[0]["value"] == 1523400
    assert metrics[0]["unit"] == "tokens/min"


def test_workspace_tuning_job(client):
    ws = at.Workspace(
        name="tune1",
        resource=at.ResourceSpec(instanceType=SKU),
        tuning=at.TuningSpec(preset=at.PresetSpec(name="llama-3-8b"),
                             method="qlora",
                             input=at.DataSource(urls=["http://d/x.json"]),
                             output=at.DataDestination(image="out:v1")))
    _store_ws(client, ws)
    client.create(_node("n1", labels={
        "node.kubernetes.io/instance-type": SKU}))
    r = _reconciler(client)
    res = r.reconcile(ws)
    assert not res.done
    job = client.get("Job", "default", "tune1")
    job["status"] = {"succeeded": 1}
    client.update(job)
    res = r.reconcile(ws)
    assert res.done and ws.status.state == "Succeeded"


# ------------------------------------------------------------ provisioners
def test_karpenter_provisioner_creates_claims(client):
    ws = _ws()
    prov = make_provisioner("karpenter", client)
    created = prov.provision_nodes(ws, 2)
    assert len(created) == 2
    claims = client.list("NodeClaim", "default")
    assert len(claims) == 2
    assert claims[0]["spec"]["requirements"][0]["values"] == [SKU]
    # idempotent
    assert prov.provision_nodes(ws, 2) == []
    # readiness: bind a node
    c = claims[0]
    c["status"] = {"nodeName": "nk1"}
    client.update(c)
    client.create(_node("nk1"))
    assert prov.ensure_nodes_ready(ws, 2) == ["nk1"]
    prov.delete_nodes(ws)
    assert client.list("NodeClaim", "default") == []


def test_byo_prefers_preferred_nodes(client):
    ws = _ws()
    ws.resource.preferredNodes = ["p2"]
    client.create(_node("p1", labels={
        "node.kubernetes.io/instance-type": SKU}))
    client.create(_node("p2", labels={
        "node.kubernetes.io/instance-type": SKU}))
    prov = BYOProvisioner(client)
    assert prov.ensure_nodes_ready(ws, 1) == ["p2"]


# ------------------------------------------------------------ inferenceset
def _iset(replicas=2):
    return at.InferenceSet("is1", spec=at.InferenceSetSpec(
        replicas=replicas, workspaceTemplate=_ws("tpl")))


def test_inferenceset_scale_up_down(client):
    r = InferenceSetReconciler(client)
    iset = _iset(3)
    res = r.reconcile(iset)
    assert res.created == 3
    names = [o["metadata"]["name"]
             for o in client.list("Workspace", "default")]
    assert names == ["is1-0", "is1-1", "is1-2"]
    iset.spec.replicas = 1
    res = r.reconcile(iset)
    assert res.deleted == 2
    assert len(client.list("Workspace", "default")) == 1


def test_inferenceset_deletes_old_revision_first(client):
    r = InferenceSetReconciler(client)
    iset = _iset(2)
    r.reconcile(iset)
    # mark is1-0 as old revision + Running, is1-1 current + Running
    for i, obj in enumerate(client.list("Workspace", "default")):
        obj["status"] = {"state": "Running"}
        if obj["metadata"]["name"] == "is1-0":
            obj["metadata"]["labels"]["inferenceset.kaito.io/revision"] = "old"
        client.update(obj)
    iset.spec.replicas = 1
    r.reconcile(iset)
    remaining = client.list("Workspace", "default")
    assert [o["metadata"]["name"] for o in remaining] == ["is1-1"]


def test_inferenceset_aggregates_tpm_and_selector(client):
    r = InferenceSetReconciler(client)
    iset = _iset(2)
    r.reconcile(iset)
    for obj in client.list("Workspace", "default"):
        obj["status"] = {"state": "Running", "performance": {
            "metrics": [{"name": "peakTokensPerMinute", "value": 100.0}]}}
        client.update(obj)
    r.reconcile(iset)
    assert iset.status.readyReplicas == 2
    assert iset.status.aggregatedPeakTokensPerMinute == 200.0
    assert iset.status.selector == "inferenceset.kaito.sh/created-by=is1"


def test_inferenceset_marks_upgrade_label(client):
    r = InferenceSetReconciler(client)
    iset = _iset(1)
    r.reconcile(iset)
    # change the template → new revision
    iset.spec.workspaceTemplate.inference.preset.name = "llama-3-70b"
    r.reconcile(iset)
    obj = client.list("Workspace", "default")[0]
    assert at.LABEL_UPGRADE_TO_VERSION in obj["metadata"]["labels"]


# ------------------------------------------------------------ modelmirror
def test_modelmirror_managed_flow(client):
    from kaito_amd.operator.controllers.modelmirror import (
        ModelMirror, ModelMirrorReconciler, PHASE_DOWNLOADING, PHASE_READY)
    logs = {"text": "downloading... 42%\n"}
    r = ModelMirrorReconciler(client,
                              get_pod_logs=lambda ns, n: logs["text"])
    mm = ModelMirror(name="llama8b", modelName="llama-3-8b")
    # first reconcile: creates PVC + job, phase pending→downloading
    assert r.reconcile(mm) == "Pending"
    assert client.get("PersistentVolumeClaim", "kaito-system",
                      "modelmirror-llama8b")
    job = client.get("Job", "kaito-system", "modelmirror-llama8b-download")
    job["status"] = {"active": 1}
    client.update(job)
    assert r.reconcile(mm) == PHASE_DOWNLOADING
    assert mm.status["progress"] == 42
    job["status"] = {"succeeded": 1}
    client.update(job)
    assert r.reconcile(mm) == PHASE_READY


def test_modelmirror_static_and_failure(client):
    from kaito_amd.operator.controllers.modelmirror import (
        ModelMirror, ModelMirrorReconciler, classify_download_failure)
    r = ModelMirrorReconciler(client)
    mm = ModelMirror(name="s", mode="Static", staticVolumePath="/weights/x")
    assert r.reconcile(mm) == "Ready"
    assert classify_download_failure("HTTP 401 Unauthorized") == "AuthFailure"
    assert classify_download_failure("model not found 404") == "ModelNotFound"
    assert classify_download_failure("no space left on device") == "OutOfDisk"


# ------------------------------------------------------------ multirole
def test_multirole_creates_role_sets_and_pool(client):
    from kaito_amd.operator.controllers.multirole import (
        MultiRoleInference, MultiRoleInferenceReconciler, RoleSpec)
    r = MultiRoleInferenceReconciler(client)
    mri = MultiRoleInference(name="pd", preset="llama-3-8b",
                             prefill=RoleSpec(replicas=2, instanceType=SKU),
                             decode=RoleSpec(replicas=3, instanceType=SKU))
    st = r.reconcile(mri)
    pre = client.get("InferenceSet", "default", "pd-prefill")
    dec = client.get("InferenceSet", "default", "pd-decode")
    assert pre["spec"]["replicas"] == 2 and dec["spec"]["replicas"] == 3
    assert pre["metadata"]["labels"][at.LABEL_INFERENCE_ROLE] == "prefill"
    pool = client.get("InferencePool", "default", "pd-pool")
    plugins = [p["name"] for p in pool["spec"]["eppConfig"]["plugins"]]
    assert "prefill-filter" in plugins and "kv-cache-utilization-scorer" in plugins
    assert not st["ready"]
    # roles become ready
    for obj in (pre, dec):
        obj["status"] = {"readyReplicas": obj["spec"]["replicas"]}
        client.update(obj)
    st = r.reconcile(mri)
    assert st["ready"]
    # scale decode
    mri.decode.replicas = 5
    r.reconcile(mri)
    assert client.get("InferenceSet", "default",
                      "pd-decode")["spec"]["replicas"] == 5


# ------------------------------------------------------------ drift/upgrade
def test_drift_serializes_remediation(client):
    from kaito_amd.operator.controllers.lifecycle import DriftReconciler
    for i in range(2):
        client.create({
            "apiVersion": "kaito.sh/v1beta1", "kind": "Workspace",
            "metadata": {"name": f"is1-{i}", "namespace": "default",
                         "labels": {at.LABEL_INFERENCESET_CREATED_BY: "is1"},
                         "annotations": {"kaito.sh/node-drifted": "true"}},
            "spec": {}, "status": {}})
    d = DriftReconciler(client)
    active = d.reconcile("is1")
    assert active == "is1-0"
    np = client.get("NodePool", "default", "is1-0-nodepool")
    assert np["spec"]["disruption"]["budgets"] == [{"nodes": "1"}]
    # second tick: still the same one (serialized)
    assert d.reconcile("is1") == "is1-0"
    # remediation completes: annotation cleared
    ws = client.get("Workspace", "default", "is1-0")
    ws["metadata"]["annotations"] = {}
    client.update(ws)
    assert d.reconcile("is1") == "is1-1"
    np = client.get("NodePool", "default", "is1-0-nodepool")
    assert np["spec"]["disruption"]["budgets"] == [{"nodes": "0"}]


def test_autoupgrade_surge_and_inplace(client):
    from kaito_amd.operator.controllers.lifecycle import AutoUpgradeRunner
    for i in range(2):
        client.create({
            "apiVersion": "kaito.sh/v1beta1", "kind": "Workspace",
            "metadata": {"name": f"is2-{i}", "namespace": "default",
                         "labels": {
                             at.LABEL_INFERENCESET_CREATED_BY: "is2",
                             "inferenceset.kaito.io/revision": "old"}},
            "spec": {}, "status": {}})
    surge = AutoUpgradeRunner(client, "v2", strategy="Surge")
    assert sorted(surge.poll("is2")) == ["is2-0", "is2-1"]
    ws = client.get("Workspace", "default", "is2-0")
    assert ws["metadata"]["labels"][at.LABEL_UPGRADE_TO_VERSION] == "v2"
    inplace = AutoUpgradeRunner(client, "v3", strategy="InPlace")
    inplace.poll("is2")
    ws = client.get("Workspace", "default", "is2-0")
    assert ws["metadata"]["labels"]["inferenceset.kaito.io/revision"] == "v3"


def test_maintenance_window():
    import time as _t
    from kaito_amd.operator.controllers.lifecycle import in_maintenance_window
    now = _t.struct_time((2026, 1, 1, 3, 0, 0, 2, 1, 0))  # 03:00, Wed
    assert in_maintenance_window("0 3 * * *", now)
    assert not in_maintenance_window("0 4 * * *", now)
    assert in_maintenance_window("", now)
    assert in_maintenance_window("0 3 * * 2", now)
    assert not in_maintenance_window("0 3 * * 5", now)


# ------------------------------------------------------------ operator loop
def test_feature_gates_parse():
    from kaito_amd.operator.featuregates import (FeatureGateError,
                                                 parse_feature_gates)
    g = parse_feature_gates("ModelMirror=true, enableMIG=false")
    assert g["ModelMirror"] and not g["enableMIG"]
    assert g["enableInferenceSetController"]  # default preserved
    with pytest.raises(FeatureGateError):
        parse_feature_gates("bogusGate=true")
    with pytest.raises(FeatureGateError):
        parse_feature_gates("ModelMirror=maybe")


def test_operator_loop_end_to_end(client):
    """CR dict in store → loop tick reconciles it to Running."""
    from kaito_amd.operator.main import OperatorLoop, workspace_from_obj
    client.create({
        "apiVersion": "kaito.sh/v1beta1", "kind": "Workspace",
        "metadata": {"name": "ws-e2e", "namespace": "default"},
        "spec": {"resource": {"instanceType": SKU},
                 "inference": {"preset": {"name": "llama-3-8b"}}},
        "status": {}})
    client.create(_node("n1", labels={
        "node.kubernetes.io/instance-type": SKU}))
    loop = OperatorLoop(client, provisioner="byo")
    assert loop.tick() == 1
    ss = client.get("StatefulSet", "default", "ws-e2e")
    ss["status"] = {"readyReplicas": 1}
    client.update(ss)
    loop.tick()
    stored = client.get("Workspace", "default", "ws-e2e")
    assert stored["status"]["state"] == "Running"
    # typed roundtrip
    ws = workspace_from_obj(stored)
    assert ws.inference.preset.name == "llama-3-8b"


def test_operator_metrics_poller():
    from kaito_amd.operator.metrics import monitor_workspaces, render
    monitor_workspaces([
        {"status": {"state": "Running"},
         "spec": {"inference": {"preset": {"name": "llama-3-8b"}}}},
        {"status": {"state": "Running"}, "spec": {}},
        {"status": {}, "spec": {}},
    ])
    body = render().decode()
    assert 'kaito_workspace_count{phase="Running"} 2.0' in body
    assert 'kaito_workspace_count{phase="Pending"} 1.0' in body
    assert 'kaito_workspace_preset_count{preset="llama-3-8b"} 1.0' in body


def test_inferenceset_gateway_manifests(client):
    from kaito_amd.operator.controllers.inferenceset import \
        InferenceSetReconciler
    r = InferenceSetReconciler(client, gateway_api=True)
    iset = _iset(1)
    r.reconcile(iset)
    oci = client.get("OCIRepository", "default", "is1-router")
    assert oci["spec"]["url"].startswith("oci://")
    hr = client.get("HelmRelease", "default", "is1-router")
    values = hr["spec"]["values"]
    ml = values["inferencePool"]["modelServers"]["matchLabels"]
    assert ml["apps.kubernetes.io/pod-index"] == "0"
    assert ml["inferenceset.kaito.sh/created-by"] == "is1"
    assert "kv-cache-utilization-scorer" in values["epp"]["plugins"]


def test_admission_webhook_allows_and_denies():
    from fastapi.testclient import TestClient
    from kaito_amd.operator.webhooks import build_webhook_app
    app = build_webhook_app(sku_handler=get_sku_handler("azure"),
                            known_presets={"llama-3-8b"})
    c = TestClient(app)

    def review(obj, group="workspace.kaito.sh"):
        return c.post(f"/validate/{group}", json={
            "request": {"uid": "u1", "object": obj}}).json()["response"]

    good = {"metadata": {"name": "w", "namespace": "default"},
            "spec": {"resource": {"instanceType": SKU},
                     "inference": {"preset": {"name": "llama-3-8b"}}}}
    assert review(good)["allowed"]
    bad = {"metadata": {"name": "w"},
           "spec": {"resource": {"instanceType": SKU},
                    "inference": {"preset": {"name": "nope"}}}}
    r = review(bad)
    assert not r["allowed"] and "nope" in r["status"]["message"]
    both = {"metadata": {"name": "w"},
            "spec": {"resource": {"instanceType": SKU},
                     "inference": {"preset": {"name": "llama-3-8b"}},
                     "tuning": {"method": "lora",
                                "input": {"urls": ["u"]},
                                "output": {"image": "i"}}}}
    assert not review(both)["allowed"]
    iset = {"metadata": {"name": "i"},
            "spec": {"replicas": 2, "workspaceTemplate": good["spec"]}}
    assert review(iset, "inferenceset.kaito.sh")["allowed"]


# ------------------------------------------------------------ ragengine
def test_ragengine_reconcile(client):
    from kaito_amd.operator.controllers.ragengine import (
        RAGEngineReconciler, rag_env_from_spec)
    rag = at.RAGEngine("rag1", spec=at.RAGEngineSpec(
        compute=at.ResourceSpec(instanceType=SKU),
        embedding={"local": {"modelID": "BAAI/bge-small-en-v1.5"}},
        inferenceService={"url": "http://ws1/v1/chat/completions",
                          "contextWindow": 4096},
        storage={"vectorDBType": "faiss"},
        guardrails={"enabled": True, "policy": "blocked_keywords: ['x']\n",
                    "hotReload": True}))
    r = RAGEngineReconciler(client)
    assert r.reconcile(rag) is False          # deployment not ready yet
    dep = client.get("Deployment", "default", "rag1")
    c = dep["spec"]["template"]["spec"]["containers"][0]
    env = {e["name"]: e["value"] for e in c["env"]}
    assert env["EMBEDDING_SOURCE_TYPE"] == "local"
    assert env["LOCAL_EMBEDDING_MODEL_ID"] == "BAAI/bge-small-en-v1.5"
    assert env["LLM_INFERENCE_URL"] == "http://ws1/v1/chat/completions"
    assert env["LLM_CONTEXT_WINDOW"] == "4096"
    assert env["OUTPUT_GUARDRAILS_ENABLED"] == "true"
    assert c["resources"]["limits"]["amd.com/gpu"] == "1"  # local embedding
    assert client.get("ConfigMap", "default", "rag1-guardrails")
    assert client.get("Service", "default", "rag1")
    # becomes ready
    dep["status"] = {"readyReplicas": 1}
    client.update(dep)
    assert r.reconcile(rag) is True
    assert rag.status["state"] == "Ready"
    conds = {c["type"]: c["status"] for c in rag.status["conditions"]}
    assert conds["ServiceReady"] == "True"


def test_ragengine_remote_embedding_no_gpu(client):
    from kaito_amd.operator.controllers.ragengine import RAGEngineReconciler
    rag = at.RAGEngine("rag2", spec=at.RAGEngineSpec(
        embedding={"remote": {"url": "http://emb/v1/embeddings"}},
        inferenceService={"url": "http://ws1"}))
    RAGEngineReconciler(client).reconcile(rag)
    c = client.get("Deployment", "default",
                   "rag2")["spec"]["template"]["spec"]["containers"][0]
    assert "resources" not in c            # no GPU for remote embedding
    env = {e["name"]: e["value"] for e in c["env"]}
    assert env["EMBEDDING_SOURCE_TYPE"] == "remote"


def test_operator_loop_reconciles_inferencesets(client):
    from kaito_amd.operator.main import OperatorLoop
    client.create({
        "apiVersion": "kaito.sh/v1beta1", "kind": "InferenceSet",
        "metadata": {"name": "is-loop", "namespace": "default"},
        "spec": {"replicas": 2, "workspaceTemplate": {
            "resource": {"instanceType": SKU},
            "inference": {"preset": {"name": "llama-3-8b"}}}},
        "status": {}})
    client.create(_node("n1", labels={
        "node.kubernetes.io/instance-type": SKU}))
    loop = OperatorLoop(client, provisioner="byo")
    loop.tick()   # creates child workspaces
    children = client.list("Workspace", "default",
                           {at.LABEL_INFERENCESET_CREATED_BY: "is-loop"})
    assert len(children) == 2
    loop.tick()   # reconciles the children to statefulsets
    assert client.get("StatefulSet", "default", "is-loop-0")


def test_readiness_timeout_scales_with_model_size():
    from kaito_amd.operator.manifests import readiness_timeout_for
    from kaito_amd.models import get_model_config
    small = readiness_timeout_for(get_model_config("tiny-llama-test"))
    mid = readiness_timeout_for(get_model_config("llama-3-8b"))
    big = readiness_timeout_for(get_model_config("llama-3-70b"))
    assert small == 600                      # floor
    assert mid == 600 or mid < big           # 8B near floor, 70B scaled
    assert big > 1200                        # ~141 GiB * 12 s


def test_controller_expectations_guard():
    import time as _t
    from kaito_amd.operator import expectations as ex
    e = ex.ControllerExpectations()
    assert e.satisfied("a")                 # nothing pending
    e.expect_creations("a", 2)
    assert not e.satisfied("a")
    e.creation_observed("a")
    assert not e.satisfied("a")
    e.creation_observed("a")
    assert e.satisfied("a")                 # fulfilled → cleared
    assert e.pending("a") is None
    e.expect_deletions("b", 1)
    assert not e.satisfied("b")
    old = ex.EXPECTATION_TIMEOUT_S
    try:
        ex.EXPECTATION_TIMEOUT_S = 0.0      # expire immediately
        _t.sleep(0.01)
        assert e.satisfied("b")             # expired → unwedged
    finally:
        ex.EXPECTATION_TIMEOUT_S = old


def test_inferenceset_expectations_block_stale_scale(monkeypatch):
    """With an artificially stale cache (creations expected but not yet
    observed), the reconciler must requeue instead of re-creating."""
    from kaito_amd.operator.kubeclient import FakeKubeClient
    from kaito_amd.operator.controllers.inferenceset import (
        InferenceSetReconciler)
    from kaito_amd.operator.main import inferenceset_from_obj
    client = FakeKubeClient()
    rec = InferenceSetReconciler(client)
    iset = inferenceset_from_obj({
        "metadata": {"name": "pool", "namespace": "default"},
        "spec": {"replicas": 2, "workspaceTemplate": {
            "resource": {"instanceType": "Standard_MI355X_v1"},
            "inference": {"preset": {"name": "llama-3-8b"}}}}})
    r1 = rec.reconcile(iset)
    assert r1.created == 2
    # simulate unobserved in-flight creations
    rec.expectations.expect_creations("default/pool", 1)
    r2 = rec.reconcile(iset)
    assert r2.created == 0 and r2.deleted == 0
    assert r2.requeue_after_s     # requeued, no scale action
    rec.expectations.creation_observed("default/pool")
    r3 = rec.reconcile(iset)
    assert r3.created == 0            # already at desired; acts normally


def test_make_kube_client_falls_back_to_fake():
    """Air-gapped image has no `kubernetes` package → factory returns the
    in-memory fake; selector helper stays deterministic."""
    from kaito_amd.operator.kubeclient import FakeKubeClient
    from kaito_amd.operator.kubeclient_incluster import (_selector_str,
                                                         make_kube_client)
    c = make_kube_client()
    assert isinstance(c, FakeKubeClient)
    assert _selector_str({"b": "2", "a": "1"}) == "a=1,b=2"
    assert _selector_str(None) is None


def test_workspace_deletion_finalizer_flow():
    """deletionTimestamp set → reconcile tears down the StatefulSet and
    Services, provisions nothing, sets WorkspaceDeleting, and drops the
    finalizer (reference: garbageCollectWorkspace)."""
    from kaito_amd.operator.api_types import (COND_WORKSPACE_DELETING,
                                              FINALIZER_WORKSPACE)
    from kaito_amd.operator.kubeclient import FakeKubeClient, NotFound
    from kaito_amd.operator.controllers.workspace import WorkspaceReconciler
    from kaito_amd.operator.main import workspace_from_obj
    from kaito_amd.operator.nodeprovision import make_provisioner
    from kaito_amd.operator.sku import get_sku_handler
    client = FakeKubeClient()
    rec = WorkspaceReconciler(client, get_sku_handler("azure"),
                              make_provisioner("byo", client),
                              "img:latest")
    obj = {"metadata": {"name": "ws1", "namespace": "default"},
           "spec": {"resource": {"instanceType": "Standard_MI355X_v1",
                                 "count": 1},
                    "inference": {"preset": {"name": "llama-3-8b"}}}}
    ws = workspace_from_obj(obj)
    rec.reconcile(ws)
    assert FINALIZER_WORKSPACE in ws.finalizers       # added on first pass
    # children may be gated on node readiness; plant them for the GC check
    for kind, nm in (("StatefulSet", "ws1"), ("Service", "ws1"),
                     ("Service", "ws1-headless")):
        try:
            client.get(kind, "default", nm)
        except NotFound:
            client.create({"kind": kind,
                           "metadata": {"name": nm,
                                        "namespace": "default"},
                           "spec": {}})

    ws.deletionTimestamp = "2026-09-12T00:00:00Z"
    rec.reconcile(ws)
    assert FINALIZER_WORKSPACE not in ws.finalizers
    assert any(c.type == COND_WORKSPACE_DELETING and c.status == "True"
               for c in ws.status.conditions)
    import pytest as _pt
    with _pt.raises(NotFound):
        client.get("StatefulSet", "default", "ws1")
    with _pt.raises(NotFound):
        client.get("Service", "default", "ws1")


def test_inferenceset_deletion_flow():
    from kaito_amd.operator.api_types import (COND_INFERENCESET_DELETING,
                                              FINALIZER_INFERENCESET)
    from kaito_amd.operator.kubeclient import FakeKubeClient
    from kaito_amd.operator.controllers.inferenceset import (
        InferenceSetReconciler)
    from kaito_amd.operator.main import inferenceset_from_obj
    client = FakeKubeClient()
    rec = InferenceSetReconciler(client)
    iset = inferenceset_from_obj({
        "metadata": {"name": "pool2", "namespace": "default"},
        "spec": {"replicas": 2, "workspaceTemplate": {
            "resource": {"instanceType": "Standard_MI355X_v1"},
            "inference": {"preset": {"name": "llama-3-8b"}}}}})
    rec.reconcile(iset)
    assert FINALIZER_INFERENCESET in iset.finalizers
    assert len(client.list("Workspace")) == 2
    iset.deletionTimestamp = "2026-09-12T00:00:00Z"
    r = rec.reconcile(iset)
    assert r.deleted == 2
    assert len(client.list("Workspace")) == 0
    assert FINALIZER_INFERENCESET not in iset.finalizers
    assert any(c.type == COND_INFERENCESET_DELETING and c.status == "True"
               for c in iset.status.conditions)


def test_ragengine_deletion_flow(client):
    from kaito_amd.operator.controllers.ragengine import RAGEngineReconciler
    from kaito_amd.operator.kubeclient import NotFound
    rag = at.RAGEngine("rag2", spec=at.RAGEngineSpec(
        compute=at.ResourceSpec(instanceType=SKU),
        embedding={"remote": {"url": "http://emb"}},
        inferenceService={"url": "http://ws1/v1/chat/completions"}))
    r = RAGEngineReconciler(client)
    r.reconcile(rag)
    assert client.get("Deployment", "default", "rag2")
    rag.deletionTimestamp = "2026-09-12T00:00:00Z"
    assert r.reconcile(rag) is False
    import pytest as _pt
    with _pt.raises(NotFound):
        client.get("Deployment", "default", "rag2")
    with _pt.raises(NotFound):
        client.get("Service", "default", "rag2")
    conds = {c["type"]: c["status"] for c in rag.status["conditions"]}
    assert conds["RAGEngineDeleting"] == "True"
