"""CRD schema tests: the generated OpenAPI v3 schemas must be real
structural schemas (not preserve-unknown stubs), stay in sync with the
generator, and accept/reject sample CRs the way the admission webhook
does (reference: controller-gen output in config/crd/bases/*.yaml +
api/v1beta1/workspace_validation.go)."""
import subprocess
import sys
from pathlib import Path

import pytest
import yaml

REPO = Path(__file__).resolve().parent.parent
BASES = REPO / "config" / "crd" / "bases"


def _load(name):
    with open(BASES / name) as f:
        return yaml.safe_load(f)


def validate(schema: dict, value, path="$"):
    """Minimal OpenAPI structural-schema validator (type / properties /
    required / items / enum / minimum / additionalProperties) — enough
    to prove the schemas constrain real CRs."""
    errors = []
    t = schema.get("type")
    if schema.get("x-kubernetes-preserve-unknown-fields"):
        return errors
    if t == "object":
        if not isinstance(value, dict):
            return [f"{path}: expected object, got {type(value).__name__}"]
        props = schema.get("properties", {})
        addl = schema.get("additionalProperties")
        for k in schema.get("required", []):
            if k not in value:
                errors.append(f"{path}.{k}: required")
        for k, v in value.items():
            if k in props:
                errors += validate(props[k], v, f"{path}.{k}")
            elif isinstance(addl, dict):
                errors += validate(addl, v, f"{path}.{k}")
            elif props and addl is None:
                errors.append(f"{path}.{k}: unknown field")
    elif t == "array":
        if not isinstance(value, list):
            return [f"{path}: expected array"]
        if "maxItems" in schema and len(value) > schema["maxItems"]:
            errors.append(f"{path}: too many items")
        for i, v in enumerate(value):
            errors += validate(schema["items"], v, f"{path}[{i}]")
    elif t == "string":
        if not isinstance(value, str):
            errors.append(f"{path}: expected string")
        elif "enum" in schema and value not in schema["enum"]:
            errors.append(f"{path}: {value!r} not in {schema['enum']}")
    elif t == "integer":
        if not isinstance(value, int) or isinstance(value, bool):
            errors.append(f"{path}: expected integer")
        elif "minimum" in schema and value < schema["minimum"]:
            errors.append(f"{path}: below minimum")
    elif t == "number":
        if not isinstance(value, (int, float)):
            errors.append(f"{path}: expected number")
    elif t == "boolean":
        if not isinstance(value, bool):
            errors.append(f"{path}: expected boolean")
    return errors


def _schema(crd, version=0):
    return crd["spec"]["versions"][version]["schema"]["openAPIV3Schema"]


ALL = ["kaito.sh_workspaces.yaml", "kaito.sh_inferencesets.yaml",
       "kaito.sh_multiroleinferences.yaml", "kaito.sh_modelmirrors.yaml",
       "kaito.sh_ragengines.yaml"]


def test_generator_in_sync():
    r = subprocess.run([sys.executable, str(REPO / "tools" / "gen_crds.py"),
                        "--check"], capture_output=True, text=True)
    assert r.returncode == 0, f"CRD drift:\n{r.stderr}"


@pytest.mark.parametrize("name", ALL)
def test_schemas_are_structural(name):
    crd = _load(name)
    s = _schema(crd)
    assert s.get("type") == "object"
    assert "x-kubernetes-preserve-unknown-fields" not in s, \
        "top-level schema is still a stub"
    assert len(s.get("properties", {})) >= 4
    ver = crd["spec"]["versions"][0]
    assert ver["subresources"] == {"status": {}}
    assert ver.get("additionalPrinterColumns"), "no printer columns"


def test_workspace_schema_accepts_valid_cr():
    s = _schema(_load("kaito.sh_workspaces.yaml"))
    cr = {
        "apiVersion": "kaito.sh/v1beta1",
        "kind": "Workspace",
        "metadata": {"name": "ws"},
        "resource": {"instanceType": "Standard_ND96isr_MI355X_v5",
                     "count": 1,
                     "partition": {"partitionType": "cpx",
                                   "partitionCount": 8}},
        "inference": {
            "preset": {"name": "llama-3-8b"},
            "adapters": [{"source": {"name": "a1", "image": "r/a:1"},
                          "strength": "0.8"}],
        },
        "status": {"state": "Running", "targetNodeCount": 1,
                   "conditions": [{"type": "ResourceReady",
                                   "status": "True"}]},
    }
    assert validate(s, cr) == []


def test_workspace_schema_rejects_bad_fields():
    s = _schema(_load("kaito.sh_workspaces.yaml"))
    bad_state = {"metadata": {}, "status": {"state": "Exploded"}}
    assert any("not in" in e for e in validate(s, bad_state))
    bad_method = {"tuning": {"method": "dpo"}}
    assert any("not in" in e for e in validate(s, bad_method))
    too_many = {"inference": {"adapters": [
        {"source": {"name": f"a{i}"}} for i in range(11)]}}
    assert any("too many" in e for e in validate(s, too_many))
    unknown = {"resource": {"instanceTypo": "x"}}
    assert any("unknown field" in e for e in validate(s, unknown))


def test_inferenceset_schema_round_trip():
    s = _schema(_load("kaito.sh_inferencesets.yaml"))
    cr = {"spec": {"replicas": 2,
                   "workspaceTemplate": {
                       "resource": {"instanceType": "mi355x"},
                       "inference": {"preset": {"name": "phi-4-mini"}}},
                   "upgradeStrategy": "Surge"},
          "status": {"readyReplicas": 2,
                     "aggregatedPeakTokensPerMinute": 1234.5}}
    assert validate(s, cr) == []
    assert any("required" in e
               for e in validate(s, {"spec": {"replicas": 1}}))
    assert any("not in" in e for e in validate(
        s, {"spec": {"workspaceTemplate": {},
                     "upgradeStrategy": "YOLO"}}))


def test_ragengine_schema_round_trip():
    s = _schema(_load("kaito.sh_ragengines.yaml"))
    cr = {"spec": {
        "embedding": {"local": {"modelID": "BAAI/bge-small-en-v1.5"}},
        "inferenceService": {"url": "http://ws:5000/v1", "contextWindow": 8192},
        "storage": {"vectorDB": "faiss"},
    }}
    assert validate(s, cr) == []
    assert any("not in" in e for e in validate(
        s, {"spec": {"storage": {"vectorDB": "pinecone"}}}))


def test_modelmirror_cluster_scoped_and_modes():
    crd = _load("kaito.sh_modelmirrors.yaml")
    assert crd["spec"]["scope"] == "Cluster"
    s = _schema(crd)
    ok = {"spec": {"modelName": "meta-llama/Llama-3.1-8B-Instruct",
                   "mode": "Managed", "storageSize": "200Gi"}}
    assert validate(s, ok) == []
    assert any("required" in e for e in validate(s, {"spec": {}}))
