"""Lease-based leader election + webhook cert bootstrap.

Reference parity: controller-runtime's leader election (manager option
LeaderElection in cmd/workspace/main.go:208-230) and the knative cert
controller that provisions the webhook serving cert + patches the
ValidatingWebhookConfiguration caBundle (pkg/workspace/webhooks).

Both are implemented against the KubeClient surface so the fake client
drives them in tests; in-cluster they operate on coordination.k8s.io
Leases and admissionregistration objects.
"""
from __future__ import annotations

import datetime
import logging
import os
import subprocess
import tempfile
from typing import Optional

from .kubeclient import KubeClient, NotFound

logger = logging.getLogger("kaito_amd.operator.leader")

LEASE_NAME = "kaito-amd-workspace-leader"


def _now() -> str:
    return datetime.datetime.now(datetime.timezone.utc).strftime(
        "%Y-%m-%dT%H:%M:%S.%fZ")


def _parse(ts: str) -> datetime.datetime:
    return datetime.datetime.strptime(ts, "%Y-%m-%dT%H:%M:%S.%fZ").replace(
        tzinfo=datetime.timezone.utc)


class LeaderElector:
    """coordination.k8s.io/v1 Lease claim/renew/steal-on-expiry."""

    def __init__(self, client: KubeClient, identity: Optional[str] = None,
                 namespace: str = "kaito-system",
                 lease_seconds: float = 15.0):
        self.client = client
        self.identity = identity or f"{os.uname().nodename}-{os.getpid()}"
        self.namespace = namespace
        self.lease_seconds = lease_seconds

    def _lease(self):
        try:
            return self.client.get("Lease", self.namespace, LEASE_NAME)
        except NotFound:
            return None

    def try_acquire(self) -> bool:
        """Acquire or renew; returns True when this process is leader."""
        lease = self._lease()
        if lease is None:
            self.client.create({
                "apiVersion": "coordination.k8s.io/v1", "kind": "Lease",
                "metadata": {"name": LEASE_NAME,
                             "namespace": self.namespace},
                "spec": {"holderIdentity": self.identity,
                         "leaseDurationSeconds": int(self.lease_seconds),
                         "renewTime": _now()},
            })
            return True
        spec = lease.get("spec", {})
        holder = spec.get("holderIdentity")
        renew = spec.get("renewTime")
        expired = True
        if renew:
            age = (datetime.datetime.now(datetime.timezone.utc)
                   - _parse(renew)).total_seconds()
            expired = age > self.lease_seconds
        if holder == self.identity or expired:
            spec["holderIdentity"] = self.identity
            spec["renewTime"] = _now()
            lease["spec"] = spec
            self.client.update(lease)
            return True
        return False


def generate_self_signed_cert(service: str = "kaito-amd-webhook",
                              namespace: str = "kaito-system",
                              out_dir: Optional[str] = None):
    """Self-signed serving cert for the webhook Service DNS name via the
    openssl CLI (no python-cryptography in the image). Returns
    (cert_pem_path, key_pem_path, ca_bundle_bytes)."""
    out_dir = out_dir or tempfile.mkdtemp(prefix="kaito-webhook-cert-")
    cn = f"{service}.{namespace}.svc"
    key = os.path.join(out_dir, "tls.key")
    crt = os.path.join(out_dir, "tls.crt")
    cnf = os.path.join(out_dir, "san.cnf")
    with open(cnf, "w") as f:
        f.write(f"""[req]
distinguished_name=dn
x509_extensions=ext
prompt=no
[dn]
CN={cn}
[ext]
subjectAltName=DNS:{cn},DNS:{cn}.cluster.local
basicConstraints=critical,CA:TRUE
keyUsage=digitalSignature,keyEncipherment,keyCertSign
extendedKeyUsage=serverAuth
""")
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", key, "-out", crt, "-days", "3650", "-config", cnf],
        check=True, capture_output=True)
    with open(crt, "rb") as f:
        ca = f.read()
    return crt, key, ca


def patch_webhook_ca_bundle(client: KubeClient, ca_pem: bytes,
                            name: str = "validation.webhook.kaito.sh") -> bool:
    """Write the CA bundle into every webhook clientConfig (the knative
    cert-controller behaviour)."""
    import base64
    try:
        vwc = client.get("ValidatingWebhookConfiguration", "", name)
    except NotFound:
        return False
    b64 = base64.b64encode(ca_pem).decode()
    for w in vwc.get("webhooks", []):
        w.setdefault("clientConfig", {})["caBundle"] = b64
    client.update(vwc)
    return True
