"""Leader election, webhook cert bootstrap, RAGEngine version
conversion, and the azure provisioner / NodePool management added in
round 2 (VERDICT coverage rows: webhooks 'no cert controller / leader
election', API types 'no v1alpha1↔v1beta1 RAGEngine conversion',
node provisioning 'no azure-gpu-provisioner / NodePool management')."""
import shutil

import pytest

from kaito_amd.operator.api_types import ResourceSpec, Workspace
from kaito_amd.operator.conversion import (convert_ragengine,
                                           ragengine_to_v1alpha1,
                                           ragengine_to_v1beta1)
from kaito_amd.operator.kubeclient import FakeKubeClient
from kaito_amd.operator.leader import (LeaderElector, generate_self_signed_cert,
                                       patch_webhook_ca_bundle)
from kaito_amd.operator.nodeprovision import (AzureGPUProvisioner,
                                              KarpenterProvisioner,
                                              make_provisioner)


# ------------------------------------------------------------- leader
def test_leader_election_single_holder_and_takeover():
    c = FakeKubeClient()
    a = LeaderElector(c, identity="a", lease_seconds=10.0)
    b = LeaderElector(c, identity="b", lease_seconds=10.0)
    assert a.try_acquire()
    assert not b.try_acquire()      # a holds a fresh lease
    assert a.try_acquire()          # renewal
    # expire the lease: b may steal
    lease = c.get("Lease", "kaito-system", "kaito-amd-workspace-leader")
    lease["spec"]["renewTime"] = "2020-01-01T00:00:00.000000Z"
    c.update(lease)
    assert b.try_acquire()
    assert not a.try_acquire()


@pytest.mark.skipif(shutil.which("openssl") is None, reason="no openssl")
def test_webhook_cert_bootstrap(tmp_path):
    crt, key, ca = generate_self_signed_cert(out_dir=str(tmp_path))
    assert b"BEGIN CERTIFICATE" in ca
    c = FakeKubeClient()
    c.create({"apiVersion": "admissionregistration.k8s.io/v1",
              "kind": "ValidatingWebhookConfiguration",
              "metadata": {"name": "validation.webhook.kaito.sh"},
              "webhooks": [{"name": "validation.workspace.kaito.sh",
                            "clientConfig": {}}]})
    assert patch_webhook_ca_bundle(c, ca)
    vwc = c.get("ValidatingWebhookConfiguration", "",
                "validation.webhook.kaito.sh")
    assert vwc["webhooks"][0]["clientConfig"]["caBundle"]


# ---------------------------------------------------------- conversion
def test_ragengine_conversion_round_trip():
    alpha = {
        "apiVersion": "kaito.sh/v1alpha1", "kind": "RAGEngine",
        "metadata": {"name": "rag"},
        "spec": {
            "compute": {"instanceType": "mi355x"},
            "embedding": {"local": {"modelID": "BAAI/bge-small-en-v1.5"}},
            "inferenceService": {"url": "http://ws:5000/v1"},
            "storage": {"persistentVolumeClaim": "rag-pvc",
                        "mountPath": "/data"},
        },
    }
    beta = ragengine_to_v1beta1(alpha)
    assert beta["apiVersion"] == "kaito.sh/v1beta1"
    # storage nests under persistentVolume in the hub version
    assert beta["spec"]["storage"] == {
        "persistentVolume": {"persistentVolumeClaim": "rag-pvc",
                             "mountPath": "/data"}}
    back = ragengine_to_v1alpha1(beta)
    assert back["spec"] == alpha["spec"]
    assert convert_ragengine(alpha, "v1beta1") == beta
    with pytest.raises(ValueError):
        convert_ragengine(alpha, "v2")


# --------------------------------------------------------- provisioners
def _ws():
    return Workspace(name="w", namespace="default",
                     resource=ResourceSpec(instanceType="mi355x-8g"))


def test_azure_provisioner_claims_and_readiness():
    c = FakeKubeClient()
    p = make_provisioner("azure", c)
    assert isinstance(p, AzureGPUProvisioner)
    ws = _ws()
    created = p.provision_nodes(ws, 2)
    assert len(created) == 2
    claims = c.list("NodeClaim", "default", {"kaito.sh/workspace": "w"})
    assert len(claims) == 2
    assert claims[0]["spec"]["nodeClassRef"]["group"] == "karpenter.azure.com"
    # fulfil one claim
    claims[0]["status"]["nodeName"] = "n0"
    c.update(claims[0])
    c.create({"apiVersion": "v1", "kind": "Node",
              "metadata": {"name": "n0"},
              "status": {"conditions": [{"type": "Ready",
                                         "status": "True"}]}})
    assert p.ensure_nodes_ready(ws, 2) == ["n0"]
    p.delete_nodes(ws)
    assert not c.list("NodeClaim", "default", {"kaito.sh/workspace": "w"})


def test_karpenter_node_pool_and_drift_budget():
    c = FakeKubeClient()
    p = KarpenterProvisioner(c)
    ws = _ws()
    pool = p.ensure_node_pool(ws, replicas=2)
    assert pool["spec"]["disruption"]["budgets"] == [{"nodes": "0"}]
    p.set_drift_remediation(ws, True)
    pool = c.get("NodePool", "default", "w-pool")
    assert pool["spec"]["disruption"]["budgets"] == [{"nodes": "1"}]
    p.set_drift_remediation(ws, False)
    pool = c.get("NodePool", "default", "w-pool")
    assert pool["spec"]["disruption"]["budgets"] == [{"nodes": "0"}]
