"""Helm chart sanity: every template renders to valid YAML with default
values (a minimal Go-template substitute — helm is not in this image),
the chart installs the generated CRDs, and RBAC covers the CRDs the
operator reconciles."""
import re
from pathlib import Path

import yaml

REPO = Path(__file__).resolve().parent.parent
CHARTS = [REPO / "charts" / "kaito-amd", REPO / "charts" / "kaito-amd-ragengine"]


def _values(chart: Path) -> dict:
    with open(chart / "values.yaml") as f:
        return yaml.safe_load(f)


def _lookup(values: dict, dotted: str):
    cur = {"Values": values, "Release": {"Namespace": "kaito-system"}}
    for part in dotted.strip(".").split("."):
        cur = cur[part]
    return cur


def render(text: str, values: dict) -> str:
    """Render the subset of Go templating these charts use:
    {{ .Values.x.y }}, {{ .Release.Namespace }}, {{ $.Release.* }} and
    {{- range tuple "a" "b" }} ... {{- end }} with {{ . }}."""
    # range blocks first
    def expand_range(m):
        items = re.findall(r'"([^"]+)"', m.group(1))
        body = m.group(2)
        out = []
        for it in items:
            b = body.replace("{{ . }}", it).replace("{{. }}", it) \
                    .replace("{{ .}}", it)
            b = re.sub(r"\{\{\s*\$\.([A-Za-z0-9_.]+)\s*\}\}",
                       lambda mm: str(_lookup(values, mm.group(1))), b)
            out.append(b)
        return "".join(out)

    text = re.sub(
        r"\{\{-?\s*range\s+tuple([^}]*)\}\}(.*?)\{\{-?\s*end\s*\}\}",
        expand_range, text, flags=re.S)
    text = re.sub(r"\{\{-?\s*\.([A-Za-z0-9_.]+)\s*-?\}\}",
                  lambda m: str(_lookup(values, m.group(1))), text)
    assert "{{" not in text, f"unrendered template bits: {text[:200]}"
    return text


def _docs(chart: Path):
    vals = _values(chart)
    for tpl in sorted((chart / "templates").glob("*.yaml")):
        for doc in yaml.safe_load_all(render(tpl.read_text(), vals)):
            if doc:
                yield tpl.name, doc


def test_templates_render_to_valid_k8s_objects():
    for chart in CHARTS:
        docs = list(_docs(chart))
        assert docs, f"{chart} has no templates"
        for name, doc in docs:
            assert "apiVersion" in doc and "kind" in doc, (name, doc)
            assert doc.get("metadata", {}).get("name"), name


def test_workspace_chart_surface():
    kinds = {d["kind"] for _, d in _docs(CHARTS[0])}
    assert {"Deployment", "ServiceAccount", "ClusterRole",
            "ClusterRoleBinding", "Service",
            "ValidatingWebhookConfiguration"} <= kinds
    # webhook covers every CRD
    for _, d in _docs(CHARTS[0]):
        if d["kind"] == "ValidatingWebhookConfiguration":
            paths = {w["clientConfig"]["service"]["path"]
                     for w in d["webhooks"]}
            assert {"/validate/workspace.kaito.sh",
                    "/validate/inferenceset.kaito.sh",
                    "/validate/ragengine.kaito.sh"} <= paths


def test_chart_installs_generated_crds():
    crds = sorted((CHARTS[0] / "crds").glob("*.yaml"))
    assert len(crds) == 5
    src = REPO / "config" / "crd" / "bases"
    for c in crds:
        assert c.read_text() == (src / c.name).read_text(), \
            f"{c.name} drifted from config/crd/bases (run tools/gen_crds.py)"


def test_rbac_covers_reconciled_resources():
    for _, d in _docs(CHARTS[0]):
        if d["kind"] == "ClusterRole":
            rules = d["rules"]
            kaito = [r for r in rules if "kaito.sh" in r["apiGroups"]]
            res = {x for r in kaito for x in r["resources"]}
            assert {"workspaces", "inferencesets", "ragengines",
                    "modelmirrors", "multiroleinferences",
                    "workspaces/status"} <= res
