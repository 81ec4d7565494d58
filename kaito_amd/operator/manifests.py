"""Kubernetes manifest generation (plain dicts → YAML).

Python re-implementation of the reference's manifest layer
(pkg/workspace/inference/preset_inferences.go:179-292 GeneratePresetInference,
pkg/workspace/manifests/manifests.go:43-200 services/statefulset,
pkg/workspace/tuning/preset_tuning.go:145 CreatePresetTuning), with the pod
spec retargeted at MI355X nodes: amd.com/gpu resources, ROCm env, our
kaito_amd.server.entrypoint command.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

from ..engine.config import ModelConfig
from .api_types import (LABEL_WORKSPACE_NAME, LABEL_WORKSPACE_NAMESPACE,
                        Workspace)
from .planner import (ParallelPlan, build_inference_command,
                      build_multinode_command, configure_parallelism)
from .sku import GPUConfig

INFERENCE_PORT = 5000
GPU_RESOURCE = "amd.com/gpu"
SHM_VOLUME = {"name": "dshm", "emptyDir": {"medium": "Memory"}}
CONFIG_MOUNT = "/mnt/config"
ADAPTER_MOUNT = "/mnt/adapter"
WEIGHTS_MOUNT = "/workspace/weights"
NVME_STORAGE_CLASS = "kaito-local-nvme-disk"


def workspace_selector(ws: Workspace) -> Dict[str, str]:
    return {LABEL_WORKSPACE_NAME: ws.name,
            LABEL_WORKSPACE_NAMESPACE: ws.namespace}


def generate_service(ws: Workspace, headless: bool = False) -> Dict[str, Any]:
    """manifests.go:43-133: ClusterIP service + headless variant for the
    multi-node rendezvous DNS."""
    name = ws.name + ("-headless" if headless else "")
    spec: Dict[str, Any] = {
        "selector": workspace_selector(ws),
        "ports": [{"name": "http", "port": 80,
                   "targetPort": INFERENCE_PORT, "protocol": "TCP"}],
    }
    if headless:
        spec["clusterIP"] = "None"
        spec["publishNotReadyAddresses"] = True
    else:
        spec["type"] = "ClusterIP"
    return {"apiVersion": "v1", "kind": "Service",
            "metadata": {"name": name, "namespace": ws.namespace,
                         "labels": workspace_selector(ws)},
            "spec": spec}


def readiness_timeout_for(model: Optional[ModelConfig]) -> int:
    """Size-scaled model readiness timeout (reference: metadata.go:54-59
    scales the startup window with TotalSafeTensorFileSize): 10 min floor
    plus ~12 s per GiB of weights (load + quantize + graph capture)."""
    if model is None:
        return 600
    gib = model.param_bytes() / (1 << 30)
    return max(600, int(gib * 12))


def _probes(readiness_timeout_s: int = 600) -> Dict[str, Any]:
    """preset_inferences.go:403-422: startup probe sized by model readiness
    timeout, liveness/readiness on /health."""
    return {
        "startupProbe": {
            "httpGet": {"path": "/health", "port": INFERENCE_PORT},
            "failureThreshold": max(readiness_timeout_s // 10, 6),
            "periodSeconds": 10},
        "livenessProbe": {
            "httpGet": {"path": "/health", "port": INFERENCE_PORT},
            "periodSeconds": 10, "failureThreshold": 6},
        "readinessProbe": {
            "httpGet": {"path": "/health", "port": INFERENCE_PORT},
            "periodSeconds": 10},
    }


def generate_inference_pod_spec(ws: Workspace, model: ModelConfig,
                                gpu: GPUConfig, plan: ParallelPlan,
                                image: str,
                                readiness_timeout_s: int = 600
                                ) -> Dict[str, Any]:
    """preset_inferences.go:522-674 GenerateInferencePodSpec for MI355X."""
    gpus = gpu.gpu_count
    if plan.num_nodes > 1:
        command = ["/bin/sh", "-c", build_multinode_command(
            model, gpu, plan, f"{ws.name}-headless.{ws.namespace}.svc")]
    else:
        command = build_inference_command(
            model, gpu, plan,
            config_file=(f"{CONFIG_MOUNT}/inference_config.yaml"
                         if ws.inference and ws.inference.config else None))
    container: Dict[str, Any] = {
        "name": ws.name,
        "image": image,
        "command": command,
        "resources": {
            "requests": {GPU_RESOURCE: str(gpus)},
            "limits": {GPU_RESOURCE: str(gpus)},
        },
        "ports": [{"containerPort": INFERENCE_PORT, "name": "http"}],
        "env": [
            {"name": "POD_INDEX", "valueFrom": {"fieldRef": {
                "fieldPath": "metadata.labels['apps.kubernetes.io/pod-index']"}}},
            {"name": "ROCM_PATH", "value": "/opt/rocm"},
            {"name": "HSA_ENABLE_IPC_MODE_LEGACY", "value": "0"},
            {"name": "KAITO_PROCESSOR", "value": "gpu"},
        ],
        "volumeMounts": [{"name": "dshm", "mountPath": "/dev/shm"}],
        **_probes(readiness_timeout_s),
    }
    volumes: List[Dict[str, Any]] = [SHM_VOLUME]
    if ws.inference and ws.inference.config:
        volumes.append({"name": "config-volume",
                        "configMap": {"name": ws.inference.config}})
        container["volumeMounts"].append(
            {"name": "config-volume", "mountPath": CONFIG_MOUNT})

    # ---- runtime-toolkit gate (the reference injects a cuda-toolkit
    # provisioner init container for JIT models, :704-738; the MI355X
    # analog asserts the node exposes gfx950 before the engine starts)
    init_containers: List[Dict[str, Any]] = [{
        "name": "rocm-runtime-check",
        "image": image,
        "command": ["/bin/sh", "-c",
                    "rocminfo | grep -q gfx950 || "
                    "{ echo 'node is not gfx950'; exit 1; }"],
        "resources": {"limits": {GPU_RESOURCE: "1"}},
    }]

    # ---- adapters (+ per-adapter strength env, :886-956, :946-951)
    for ad in (ws.inference.adapters if ws.inference else []):
        nm = ad.source.get("name", "adapter")
        init_containers.append({
            "name": f"adapter-{nm}",
            "image": ad.source.get("image", ""),
            "command": ["/bin/sh", "-c",
                        f"mkdir -p {ADAPTER_MOUNT}/{nm} && "
                        f"cp -r /data/* {ADAPTER_MOUNT}/{nm}/"],
            "volumeMounts": [{"name": "adapter-volume",
                              "mountPath": ADAPTER_MOUNT}],
        })
        if ad.strength:
            container["env"].append({
                "name": f"KAITO_ADAPTER_STRENGTH_{nm.upper().replace('-', '_')}",
                "value": str(ad.strength)})
    if len(init_containers) > 1:
        volumes.append({"name": "adapter-volume", "emptyDir": {}})
        container["volumeMounts"].append(
            {"name": "adapter-volume", "mountPath": ADAPTER_MOUNT})

    # ---- local-weights NVMe cache + download monitor (:158-177; the
    # entrypoint serves kaito_model_download_* gauges on /metrics while
    # weights stream in — server/download_monitor.py)
    from .api_types import (ANNOTATION_DISABLE_BENCHMARK,
                            ANNOTATION_USE_LOCAL_WEIGHTS,
                            LABEL_INFERENCE_ROLE)
    use_local = ws.annotations.get(ANNOTATION_USE_LOCAL_WEIGHTS) == "true"
    if use_local:
        container["volumeMounts"].append(
            {"name": "weights-cache", "mountPath": WEIGHTS_MOUNT})
        container["env"] += [
            {"name": "KAITO_WEIGHTS_PATH", "value": WEIGHTS_MOUNT},
            {"name": "KAITO_DOWNLOAD_MONITOR", "value": "1"},
        ]

    # ---- benchmark startup probe (:455-480): one-shot saturation
    # benchmark emitting KAITO_BENCHMARK_RESULT to the pod log, read
    # back by the controller (controllers/workspace.py _ingest_benchmark)
    if ws.annotations.get(ANNOTATION_DISABLE_BENCHMARK) != "true":
        container["startupProbe"] = {
            "exec": {"command": [
                "python3", "-m", "kaito_amd.server.benchmark_entrypoint",
                "--once"]},
            "failureThreshold": max(readiness_timeout_s // 10, 6),
            "periodSeconds": 10, "timeoutSeconds": 600,
        }

    # ---- P/D disaggregation (decode role: engine on :5001 behind the
    # routing sidecar on :5000; KV-transfer side channel env —
    # preset_inferences.go:1082-1152)
    role = ws.labels.get(LABEL_INFERENCE_ROLE)
    sidecars: List[Dict[str, Any]] = []
    if role:
        container["env"].append(
            {"name": "KAITO_INFERENCE_ROLE", "value": role})
        container["env"].append(
            {"name": "KAITO_KV_TRANSFER_PORT", "value": "5600"})
    if role == "decode":
        container["env"].append(
            {"name": "KAITO_INFERENCE_PORT", "value": "5001"})
        container["ports"] = [
            {"containerPort": 5001, "name": "engine"}]
        sidecars.append({
            "name": "routing-proxy",
            "image": image,
            "command": ["python3", "-m", "kaito_amd.server.dp_frontend",
                        "--port", "5000", "--backends",
                        "http://127.0.0.1:5001"],
            "ports": [{"containerPort": INFERENCE_PORT, "name": "http"}],
        })

    spec = {
        "containers": [container] + sidecars,
        "initContainers": init_containers,
        "volumes": volumes,
        "tolerations": [
            {"key": "sku", "operator": "Equal", "value": "gpu",
             "effect": "NoSchedule"},
            {"key": GPU_RESOURCE, "operator": "Exists",
             "effect": "NoSchedule"},
        ],
    }
    if ws.resource.instanceType:
        spec["nodeSelector"] = {
            "node.kubernetes.io/instance-type": ws.resource.instanceType}
    return spec


def generate_statefulset(ws: Workspace, model: ModelConfig, gpu: GPUConfig,
                         image: str,
                         plan: Optional[ParallelPlan] = None
                         ) -> Dict[str, Any]:
    """manifests.go:135-200 GenerateStatefulSetManifest."""
    plan = plan or configure_parallelism(model, gpu)
    sel = workspace_selector(ws)
    pod_spec = generate_inference_pod_spec(
        ws, model, gpu, plan, image,
        readiness_timeout_s=readiness_timeout_for(model))
    sts = {
        "apiVersion": "apps/v1",
        "kind": "StatefulSet",
        "metadata": {"name": ws.name, "namespace": ws.namespace,
                     "labels": sel},
        "spec": {
            "replicas": plan.num_nodes,
            "selector": {"matchLabels": sel},
            "serviceName": ws.name + "-headless",
            "podManagementPolicy": "Parallel",
            "template": {
                "metadata": {"labels": sel},
                "spec": pod_spec,
            },
        },
    }
    # local NVMe weights cache: per-pod PVC template on the NVMe storage
    # class (preset_inferences.go:158-177, :262-268) when the workspace
    # opts into local weights
    from .api_types import ANNOTATION_USE_LOCAL_WEIGHTS
    if ws.annotations.get(ANNOTATION_USE_LOCAL_WEIGHTS) == "true":
        size_gib = max(64, int(model.param_bytes() / (1 << 30) * 2.5) + 48)
        sts["spec"]["volumeClaimTemplates"] = [{
            "metadata": {"name": "weights-cache"},
            "spec": {
                "accessModes": ["ReadWriteOnce"],
                "storageClassName": NVME_STORAGE_CLASS,
                "resources": {"requests": {
                    "storage": f"{(size_gib + 9) // 10 * 10}Gi"}},
            },
        }]
    return sts


def generate_tuning_job(ws: Workspace, model: ModelConfig, gpu: GPUConfig,
                        image: str) -> Dict[str, Any]:
    """preset_tuning.go:145 CreatePresetTuning → batch/v1 Job with
    data-source init container and result output volume."""
    assert ws.tuning is not None
    sel = workspace_selector(ws)
    data_init: Dict[str, Any] = {
        "name": "data-downloader", "image": "busybox",
        "command": ["sh", "-c", " && ".join(
            f"wget -O /mnt/data/{i}.dat {u}"
            for i, u in enumerate(ws.tuning.input.urls or ["none"]))],
        "volumeMounts": [{"name": "data-volume", "mountPath": "/mnt/data"}],
    }
    container = {
        "name": ws.name,
        "image": image,
        "command": ["python3", "-m", "kaito_amd.tuning.fine_tuning",
                    "--model", model.name,
                    "--method", ws.tuning.method,
                    "--num-processes", str(gpu.gpu_count)],
        "resources": {"requests": {GPU_RESOURCE: str(gpu.gpu_count)},
                      "limits": {GPU_RESOURCE: str(gpu.gpu_count)}},
        "volumeMounts": [
            {"name": "data-volume", "mountPath": "/mnt/data"},
            {"name": "results-volume", "mountPath": "/mnt/results"},
            {"name": "dshm", "mountPath": "/dev/shm"},
        ],
    }
    return {
        "apiVersion": "batch/v1",
        "kind": "Job",
        "metadata": {"name": ws.name, "namespace": ws.namespace, "labels": sel},
        "spec": {
            "backoffLimit": 2,
            "template": {"metadata": {"labels": sel}, "spec": {
                "restartPolicy": "Never",
                "initContainers": [data_init],
                "containers": [container],
                "volumes": [
                    {"name": "data-volume", "emptyDir": {}},
                    {"name": "results-volume", "emptyDir": {}},
                    SHM_VOLUME,
                ],
            }},
        },
    }


def generate_inference_pool_oci_repository(iset_name: str, namespace: str,
                                           chart_url: str =
                                           "oci://ghcr.io/llm-d/charts"
                                           ) -> Dict[str, Any]:
    """Reference parity: GenerateInferencePoolOCIRepository
    (pkg/workspace/manifests/manifests.go:393) — Flux source for the
    llm-d router chart."""
    return {
        "apiVersion": "source.toolkit.fluxcd.io/v1",
        "kind": "OCIRepository",
        "metadata": {"name": f"{iset_name}-router", "namespace": namespace},
        "spec": {
            "interval": "10m",
            "url": chart_url,
            "ref": {"tag": "latest"},
        },
    }


def generate_inference_pool_helm_release(iset_name: str, namespace: str,
                                         epp_image: str =
                                         "ghcr.io/kaito-amd/epp:latest"
                                         ) -> Dict[str, Any]:
    """Reference parity: GenerateInferencePoolHelmRelease (manifests.go:421)
    — InferencePool + EPP with KVCache-aware routing; model servers matched
    by the InferenceSet child label + pod-index=0 (EPP pin, :431)."""
    from .api_types import LABEL_INFERENCESET_CREATED_BY
    return {
        "apiVersion": "helm.toolkit.fluxcd.io/v2",
        "kind": "HelmRelease",
        "metadata": {"name": f"{iset_name}-router", "namespace": namespace},
        "spec": {
            "interval": "10m",
            "chartRef": {"kind": "OCIRepository",
                         "name": f"{iset_name}-router"},
            "values": {
                "inferencePool": {
                    "targetPort": INFERENCE_PORT,
                    "modelServers": {"matchLabels": {
                        LABEL_INFERENCESET_CREATED_BY: iset_name,
                        "apps.kubernetes.io/pod-index": "0",
                    }},
                },
                "epp": {
                    "image": epp_image,
                    "args": ["--kv-cache-events-port", "5557"],
                    "plugins": ["load-aware-scorer",
                                "kv-cache-utilization-scorer",
                                "prefix-cache-scorer"],
                },
            },
        },
    }
