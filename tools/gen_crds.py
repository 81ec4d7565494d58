#!/usr/bin/env python3
"""Generate the CRD OpenAPI v3 schemas from the operator's API types.

The round-1 CRD YAMLs were `x-kubernetes-preserve-unknown-fields` stubs;
this generator emits full structural schemas (the reference ships
controller-gen output, config/crd/bases/kaito.sh_workspaces.yaml) for
the v1beta1/v1alpha1 surface the Python operator actually implements
(kaito_amd/operator/api_types.py, controllers/multirole.py,
controllers/modelmirror.py). Printer columns and status subresources
mirror the reference's kubectl UX.

Usage: python tools/gen_crds.py [--check]
  --check: exit 1 if the committed YAML differs from the generated one
  (drift guard, run by tests/test_crds.py).
"""
from __future__ import annotations

import argparse
import sys
from pathlib import Path

import yaml

_REPO = Path(__file__).resolve().parent.parent
OUT_DIR = _REPO / "config" / "crd" / "bases"
# Helm installs the same CRDs from the chart's crds/ directory
CHART_CRD_DIR = _REPO / "charts" / "kaito-amd" / "crds"


# ---------------------------------------------------------------- helpers
def obj(props: dict, required=None, description: str = "") -> dict:
    s = {"type": "object", "properties": props}
    if required:
        s["required"] = list(required)
    if description:
        s["description"] = description
    return s


def arr(items: dict, max_items: int | None = None) -> dict:
    s = {"type": "array", "items": items}
    if max_items is not None:
        s["maxItems"] = max_items
    return s


def s_str(enum=None, description: str = "") -> dict:
    s = {"type": "string"}
    if enum:
        s["enum"] = list(enum)
    if description:
        s["description"] = description
    return s


def s_int(minimum=None) -> dict:
    s = {"type": "integer"}
    if minimum is not None:
        s["minimum"] = minimum
    return s


def s_map_str() -> dict:
    return {"type": "object", "additionalProperties": {"type": "string"}}


def s_any(description: str = "") -> dict:
    s = {"type": "object", "x-kubernetes-preserve-unknown-fields": True}
    if description:
        s["description"] = description
    return s


CONDITION = obj({
    "type": s_str(),
    "status": s_str(enum=["True", "False", "Unknown"]),
    "reason": s_str(),
    "message": s_str(),
    "lastTransitionTime": s_str(),
    "observedGeneration": s_int(),
}, required=["type", "status"])


# ------------------------------------------------------------ sub-schemas
LABEL_SELECTOR = obj({
    "matchLabels": s_map_str(),
    "matchExpressions": arr(obj({
        "key": s_str(),
        "operator": s_str(enum=["In", "NotIn", "Exists", "DoesNotExist"]),
        "values": arr(s_str()),
    }, required=["key", "operator"])),
})

PARTITION = obj({
    "partitionType": s_str(
        description="GPU partition profile (MI355X: spx = whole GPU, "
                    "cpx = one partition per XCD, 36 GiB HBM3E each)"),
    "partitionCount": s_int(minimum=1),
})

RESOURCE = obj({
    "instanceType": s_str(
        description="GPU SKU to provision (e.g. an MI355X 8-GPU node "
                    "type); BYO nodes use labelSelector instead"),
    "labelSelector": LABEL_SELECTOR,
    "preferredNodes": arr(s_str()),
    "count": s_int(minimum=1),
    "partition": PARTITION,
}, description="Node/GPU requirements; the controller estimates the "
               "node count from the model memory footprint")

PRESET = obj({
    "name": s_str(description="model preset name from the catalog"),
    "presetOptions": obj({
        "image": s_str(),
        "imagePullSecrets": arr(s_str()),
        "modelAccessSecret": s_str(),
        "modelAccessMode": s_str(),
    }),
    "accessMode": s_str(enum=["public", "private"]),
}, required=["name"])

ADAPTER = obj({
    "source": obj({
        "name": s_str(),
        "image": s_str(),
        "imagePullSecrets": arr(s_str()),
    }, required=["name"]),
    "strength": s_str(description="float in (0,1] as a string"),
})

INFERENCE = obj({
    "preset": PRESET,
    "template": s_any("raw pod template for non-preset workloads"),
    "config": s_str(description="name of an inference_config.yaml "
                                "ConfigMap merged into engine args"),
    "adapters": arr(ADAPTER, max_items=10),
})

VOLUME_SOURCE = s_any("k8s VolumeSource")

TUNING = obj({
    "preset": PRESET,
    "method": s_str(enum=["lora", "qlora"]),
    "config": s_str(),
    "input": obj({
        "name": s_str(),
        "urls": arr(s_str()),
        "image": s_str(),
        "imagePullSecrets": arr(s_str()),
        "volumeSource": VOLUME_SOURCE,
    }),
    "output": obj({
        "image": s_str(),
        "imagePushSecret": s_str(),
        "volumeSource": VOLUME_SOURCE,
    }),
})

WORKSPACE_STATUS = obj({
    "conditions": arr(CONDITION),
    "workerNodes": arr(s_str()),
    "targetNodeCount": s_int(),
    "state": s_str(enum=["Pending", "Ready", "NotReady", "Running",
                         "Succeeded", "Failed"]),
    "performance": obj({
        "metrics": arr(obj({
            "name": s_str(),
            "value": {"type": "number"},
            "unit": s_str(),
            "description": s_str(),
            "config": s_map_str(),
        })),
    }),
})

# Workspace keeps the reference's TOP-LEVEL resource/inference/tuning
# layout (no .spec wrapper — workspace_types.go embeds them directly).
WORKSPACE_SCHEMA = obj({
    "apiVersion": s_str(),
    "kind": s_str(),
    "metadata": {"type": "object"},
    "resource": RESOURCE,
    "inference": INFERENCE,
    "tuning": TUNING,
    "status": WORKSPACE_STATUS,
})

INFERENCESET_SCHEMA = obj({
    "apiVersion": s_str(),
    "kind": s_str(),
    "metadata": {"type": "object"},
    "spec": obj({
        "replicas": s_int(minimum=0),
        "workspaceTemplate": obj({
            "resource": RESOURCE,
            "inference": INFERENCE,
        }),
        "upgradeStrategy": s_str(enum=["Surge", "InPlace"]),
        "maintenanceWindow": s_str(description="cron window for "
                                               "auto-upgrade"),
    }, required=["workspaceTemplate"]),
    "status": obj({
        "replicas": s_int(),
        "readyReplicas": s_int(),
        "selector": s_str(),
        "aggregatedPeakTokensPerMinute": {"type": "number"},
        "conditions": arr(CONDITION),
    }),
})

MRI_ROLE = obj({
    "replicas": s_int(minimum=1),
    "instanceType": s_str(),
})

MRI_SCHEMA = obj({
    "apiVersion": s_str(),
    "kind": s_str(),
    "metadata": {"type": "object"},
    "spec": obj({
        "preset": s_str(),
        "prefill": MRI_ROLE,
        "decode": MRI_ROLE,
    }, required=["preset"]),
    "status": obj({
        "conditions": arr(CONDITION),
        "prefillReady": s_int(),
        "decodeReady": s_int(),
    }),
})

MODELMIRROR_SCHEMA = obj({
    "apiVersion": s_str(),
    "kind": s_str(),
    "metadata": {"type": "object"},
    "spec": obj({
        "modelName": s_str(),
        "mode": s_str(enum=["Managed", "Static"]),
        "storageClassName": s_str(),
        "storageSize": s_str(),
        "staticVolumePath": s_str(),
        "namespace": s_str(),
    }, required=["modelName"]),
    "status": obj({
        "phase": s_str(),
        "progress": s_str(),
        "conditions": arr(CONDITION),
    }),
})

RAGENGINE_SPEC = obj({
    "compute": RESOURCE,
    "embedding": obj({
        "local": obj({"modelID": s_str(), "modelAccessSecret": s_str()}),
        "remote": obj({"url": s_str(), "accessSecret": s_str()}),
    }),
    "inferenceService": obj({
        "url": s_str(),
        "accessSecret": s_str(),
        "contextWindow": s_int(minimum=1),
    }),
    "storage": obj({
        "vectorDB": s_str(enum=["faiss", "qdrant"]),
        "url": s_str(),
        "accessSecret": s_str(),
        "persistentVolumeClaim": s_str(),
    }),
    "guardrails": obj({
        "enabled": {"type": "boolean"},
        "policyConfigMap": s_str(),
        "hotReload": {"type": "boolean"},
    }),
    "indexServiceName": s_str(),
    "queryServiceName": s_str(),
})

RAGENGINE_SCHEMA = obj({
    "apiVersion": s_str(),
    "kind": s_str(),
    "metadata": {"type": "object"},
    "spec": RAGENGINE_SPEC,
    "status": obj({
        "conditions": arr(CONDITION),
        "state": s_str(),
    }),
})


def crd(plural: str, kind: str, version: str, schema: dict,
        scope: str = "Namespaced", columns=None, extra_versions=None) -> dict:
    ver = {
        "name": version,
        "served": True,
        "storage": True,
        "schema": {"openAPIV3Schema": schema},
        "subresources": {"status": {}},
    }
    if columns:
        ver["additionalPrinterColumns"] = columns
    versions = [ver] + (extra_versions or [])
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"{plural}.kaito.sh"},
        "spec": {
            "group": "kaito.sh",
            "names": {"kind": kind, "listKind": f"{kind}List",
                      "plural": plural, "singular": kind.lower()},
            "scope": scope,
            "versions": versions,
        },
    }


def age_col():
    return {"name": "Age", "type": "date", "jsonPath":
            ".metadata.creationTimestamp"}


def build() -> dict:
    """filename → CRD object."""
    return {
        "kaito.sh_workspaces.yaml": crd(
            "workspaces", "Workspace", "v1beta1", WORKSPACE_SCHEMA,
            columns=[
                {"name": "Instance", "type": "string",
                 "jsonPath": ".resource.instanceType"},
                {"name": "ResourceReady", "type": "string",
                 "jsonPath":
                 ".status.conditions[?(@.type==\"ResourceReady\")].status"},
                {"name": "InferenceReady", "type": "string",
                 "jsonPath":
                 ".status.conditions[?(@.type==\"InferenceReady\")].status"},
                {"name": "JobStarted", "type": "string",
                 "jsonPath":
                 ".status.conditions[?(@.type==\"JobStarted\")].status"},
                {"name": "WorkspaceSucceeded", "type": "string",
                 "jsonPath":
                 ".status.conditions[?(@.type==\"WorkspaceSucceeded\")]"
                 ".status"},
                age_col(),
            ]),
        "kaito.sh_inferencesets.yaml": crd(
            "inferencesets", "InferenceSet", "v1beta1", INFERENCESET_SCHEMA,
            columns=[
                {"name": "Replicas", "type": "integer",
                 "jsonPath": ".spec.replicas"},
                {"name": "Ready", "type": "integer",
                 "jsonPath": ".status.readyReplicas"},
                {"name": "TPM", "type": "number",
                 "jsonPath": ".status.aggregatedPeakTokensPerMinute"},
                age_col(),
            ]),
        "kaito.sh_multiroleinferences.yaml": crd(
            "multiroleinferences", "MultiRoleInference", "v1alpha1",
            MRI_SCHEMA,
            columns=[
                {"name": "Preset", "type": "string",
                 "jsonPath": ".spec.preset"},
                {"name": "PrefillReady", "type": "integer",
                 "jsonPath": ".status.prefillReady"},
                {"name": "DecodeReady", "type": "integer",
                 "jsonPath": ".status.decodeReady"},
                age_col(),
            ]),
        "kaito.sh_modelmirrors.yaml": crd(
            "modelmirrors", "ModelMirror", "v1alpha1", MODELMIRROR_SCHEMA,
            scope="Cluster",
            columns=[
                {"name": "Model", "type": "string",
                 "jsonPath": ".spec.modelName"},
                {"name": "Mode", "type": "string",
                 "jsonPath": ".spec.mode"},
                {"name": "Phase", "type": "string",
                 "jsonPath": ".status.phase"},
                age_col(),
            ]),
        "kaito.sh_ragengines.yaml": crd(
            "ragengines", "RAGEngine", "v1alpha1", RAGENGINE_SCHEMA,
            columns=[
                {"name": "ServiceReady", "type": "string",
                 "jsonPath":
                 ".status.conditions[?(@.type==\"ServiceReady\")].status"},
                age_col(),
            ]),
    }


HEADER = ("# Generated by tools/gen_crds.py from "
          "kaito_amd/operator/api_types.py — do not edit by hand.\n")


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--check", action="store_true")
    args = p.parse_args()
    rc = 0
    OUT_DIR.mkdir(parents=True, exist_ok=True)
    CHART_CRD_DIR.mkdir(parents=True, exist_ok=True)
    for fname, doc in build().items():
        text = HEADER + yaml.safe_dump(doc, sort_keys=False, width=78)
        for path in (OUT_DIR / fname, CHART_CRD_DIR / fname):
            if args.check:
                if not path.exists() or path.read_text() != text:
                    print(f"DRIFT: {path}", file=sys.stderr)
                    rc = 1
            else:
                path.write_text(text)
                print(f"wrote {path}")
    return rc


if __name__ == "__main__":
    sys.exit(main())
