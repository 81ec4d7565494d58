"""Workspace reconciler — the Python re-implementation of the reference's
pkg/workspace/controllers/workspace_controller.go (1570 L) core path
(SURVEY.md §3.2): validate → targetNodeCount estimate → provision →
ensure Services → StatefulSet (inference) / Job (tuning) → status sync with
pod-failure classification → benchmark result ingestion.
"""
from __future__ import annotations

import hashlib
import json
import time
from dataclasses import dataclass
from typing import Callable, Dict, Optional

from ..api_types import (ANNOTATION_DISABLE_BENCHMARK,
                         ANNOTATION_WORKSPACE_HASH, COND_INFERENCE_READY,
                         COND_JOB_STARTED, COND_NODECLAIM_READY,
                         COND_NODES_READY, COND_RESOURCE_READY,
                         COND_WORKSPACE_DELETING, COND_WORKSPACE_SUCCEEDED,
                         Condition, FINALIZER_WORKSPACE,
                         LABEL_WORKSPACE_NAME, ValidationError, Workspace)
from ..estimator import NodeEstimateRequest, estimate_node_count
from ..kubeclient import KubeClient, NotFound
from ..manifests import (generate_service, generate_statefulset,
                         generate_tuning_job)
from ..nodeprovision import NodeProvisioner
from ..planner import configure_parallelism
from ..sku import CloudSKUHandler, GPUConfig
from ...models.registry import get_model_config

BENCHMARK_RESULT_TAG = "KAITO_BENCHMARK_RESULT"
BENCHMARK_CONFIG_TAG = "KAITO_BENCHMARK_CONFIG"


@dataclass
class ReconcileResult:
    requeue_after_s: float = 0.0
    done: bool = True

    @property
    def requeue(self) -> bool:
        return self.requeue_after_s > 0


def _cond(status: "list", ctype: str, ok: bool, reason: str = "",
          message: str = "") -> None:
    for c in status:
        if c.type == ctype:
            c.status = "True" if ok else "False"
            c.reason = reason
            c.message = message
            c.lastTransitionTime = time.strftime("%Y-%m-%dT%H:%M:%SZ")
            return
    status.append(Condition(ctype, "True" if ok else "False", reason, message,
                            time.strftime("%Y-%m-%dT%H:%M:%SZ")))


def classify_pod_failure(pod: Dict) -> Optional[str]:
    """Reference parity: classifyInferencePodFailure
    (workspace_controller.go:934-962) — typed condition reasons."""
    st = pod.get("status", {})
    if st.get("phase") == "Pending":
        for c in st.get("conditions", []):
            if c.get("type") == "PodScheduled" and c.get("status") == "False":
                return "Unschedulable"
    for cs in st.get("containerStatuses", []):
        waiting = cs.get("state", {}).get("waiting", {})
        terminated = cs.get("state", {}).get("terminated", {})
        last_term = cs.get("lastState", {}).get("terminated", {})
        reason = waiting.get("reason", "")
        if reason in ("ErrImagePull", "ImagePullBackOff"):
            return "ImagePullFailure"
        if reason == "CrashLoopBackOff":
            if last_term.get("reason") == "OOMKilled":
                return "OOMKilled"
            return "CrashLoop"
        if terminated.get("reason") == "OOMKilled" or \
                last_term.get("reason") == "OOMKilled":
            return "OOMKilled"
    if st.get("reason") == "Evicted":
        return "Evicted"
    return None


class WorkspaceReconciler:
    def __init__(self, client: KubeClient, sku_handler: CloudSKUHandler,
                 provisioner: NodeProvisioner,
                 image: str = "ghcr.io/kaito-amd/engine:latest",
                 get_pod_logs: Optional[Callable[[str, str], str]] = None):
        self.client = client
        self.sku = sku_handler
        self.provisioner = provisioner
        self.image = image
        self.get_pod_logs = get_pod_logs or (lambda ns, name: "")

    # ------------------------------------------------------------ helpers
    def _gpu_config(self, ws: Workspace) -> GPUConfig:
        cfg = self.sku.get_gpu_config(ws.resource.instanceType)
        if cfg is None:
            from ..sku import MI355X
            cfg = GPUConfig(ws.resource.instanceType or "byo", 8, 288, MI355X,
                            "gfx950", 7)
        # partition spec rescale (reference: ScaleGPUConfigToCount,
        # pkg/sku/helpers.go:123-132): when the workspace requests a
        # partitioned view (CPX), the estimator must see partition-sized
        # devices, capped at the requested partitionCount per node
        from ..partition import partitioned_gpu_config, validate_partition
        prof = validate_partition(ws.resource.partition, cfg)
        if prof is not None:
            cfg = partitioned_gpu_config(cfg, prof)
            pc = ws.resource.partition.partitionCount
            if pc and pc < cfg.gpu_count:
                cfg = cfg.scale_to_count(pc)
        return cfg

    def _spec_hash(self, ws: Workspace) -> str:
        payload = json.dumps({
            "inference": ws.inference.preset.name
            if ws.inference and ws.inference.preset else None,
            "tuning": ws.tuning.method if ws.tuning else None,
            "instanceType": ws.resource.instanceType,
            "count": ws.resource.count,
        }, sort_keys=True)
        return hashlib.sha256(payload.encode()).hexdigest()[:16]

    # ------------------------------------------------------------ reconcile
    def reconcile(self, ws: Workspace) -> ReconcileResult:
        conds = ws.status.conditions
        if ws.deletionTimestamp:
            # deletion flow (reference: Reconcile :117-127 →
            # garbageCollectWorkspace): tear down children, then drop the
            # finalizer so the API server can remove the object
            return self._finalize(ws)
        if FINALIZER_WORKSPACE not in ws.finalizers:
            ws.finalizers.append(FINALIZER_WORKSPACE)
        try:
            ws.validate(sku_handler=None)
        except ValidationError as e:
            _cond(conds, COND_RESOURCE_READY, False, "ValidationFailed", str(e))
            ws.status.state = "Failed"
            return ReconcileResult()

        # 1. target node count (UpdateWorkspaceTargetNodeCount :1481)
        preset = (ws.inference.preset.name if ws.inference and
                  ws.inference.preset else
                  ws.tuning.preset.name if ws.tuning and ws.tuning.preset
                  else None)
        gpu = self._gpu_config(ws)
        if preset:
            model = get_model_config(preset)
            est = estimate_node_count(NodeEstimateRequest(
                model=model, gpu=gpu, replicas=1))
            target = max(est.nodes_per_replica, ws.resource.count or 1)
        else:
            model = None
            target = ws.resource.count or 1
        ws.status.targetNodeCount = target

        # 2. provision + wait for nodes (reconcileNodes :327)
        self.provisioner.provision_nodes(ws, target)
        ready_nodes = self.provisioner.ensure_nodes_ready(ws, target)
        _cond(conds, COND_NODECLAIM_READY, True, "Provisioned")
        if len(ready_nodes) < target:
            _cond(conds, COND_NODES_READY, False, "WaitingForNodes",
                  f"{len(ready_nodes)}/{target} nodes ready")
            ws.status.state = "Pending"
            self._push_status(ws)
            return ReconcileResult(requeue_after_s=2.0, done=False)
        _cond(conds, COND_NODES_READY, True, "NodesReady")
        ws.status.workerNodes = ready_nodes
        _cond(conds, COND_RESOURCE_READY, True, "Ready")

        # 3. workload
        if ws.inference is not None:
            res = self._apply_inference(ws, model, gpu, target)
        else:
            res = self._apply_tuning(ws, model, gpu)
        self._push_status(ws)
        return res

    # ------------------------------------------------------------ inference
    def _finalize(self, ws: Workspace) -> ReconcileResult:
        _cond(ws.status.conditions, COND_WORKSPACE_DELETING, True,
              "Deleting", "tearing down child resources")
        for kind in ("StatefulSet", "Job"):
            try:
                self.client.delete(kind, ws.namespace, ws.name)
            except NotFound:
                pass
        for svc in (ws.name, f"{ws.name}-headless"):
            try:
                self.client.delete("Service", ws.namespace, svc)
            except NotFound:
                pass
        self.provisioner.delete_nodes(ws)
        if FINALIZER_WORKSPACE in ws.finalizers:
            ws.finalizers.remove(FINALIZER_WORKSPACE)
        self._push_status(ws)
        return ReconcileResult()

    def _apply_inference(self, ws: Workspace, model, gpu,
                         target: int) -> ReconcileResult:
        conds = ws.status.conditions
        self.client.apply(generate_service(ws))
        self.client.apply(generate_service(ws, headless=True))
        plan = configure_parallelism(model, gpu, num_nodes=target) \
            if model else None
        ss = generate_statefulset(ws, model, gpu, self.image, plan)
        ss["metadata"].setdefault("annotations", {})[
            ANNOTATION_WORKSPACE_HASH] = self._spec_hash(ws)
        self.client.apply(ss)

        # status from statefulset + pods
        try:
            live = self.client.get("StatefulSet", ws.namespace, ws.name)
        except NotFound:
            live = ss
        ready = live.get("status", {}).get("readyReplicas", 0)
        desired = live["spec"]["replicas"]
        pods = self.client.list("Pod", ws.namespace, {
            LABEL_WORKSPACE_NAME: ws.name})
        failure = None
        for pod in pods:
            failure = classify_pod_failure(pod)
            if failure:
                break
        if failure:
            _cond(conds, COND_INFERENCE_READY, False, failure)
            ws.status.state = "Failed" if failure in (
                "OOMKilled", "ImagePullFailure") else "NotReady"
            return ReconcileResult(requeue_after_s=10.0, done=False)
        if ready >= desired:
            _cond(conds, COND_INFERENCE_READY, True, "InferenceReady")
            ws.status.state = "Running"
            self._ingest_benchmark(ws)
            return ReconcileResult()
        _cond(conds, COND_INFERENCE_READY, False, "WaitingForPods",
              f"{ready}/{desired} ready")
        ws.status.state = "NotReady"
        return ReconcileResult(requeue_after_s=5.0, done=False)

    # ------------------------------------------------------------ tuning
    def _apply_tuning(self, ws: Workspace, model, gpu) -> ReconcileResult:
        conds = ws.status.conditions
        job = generate_tuning_job(ws, model, gpu, self.image)
        self.client.apply(job)
        try:
            live = self.client.get("Job", ws.namespace, ws.name)
        except NotFound:
            live = job
        st = live.get("status", {})
        _cond(conds, COND_JOB_STARTED, True, "JobCreated")
        if st.get("succeeded"):
            _cond(conds, COND_WORKSPACE_SUCCEEDED, True, "TuningComplete")
            ws.status.state = "Succeeded"
            return ReconcileResult()
        if st.get("failed", 0) > 2:
            _cond(conds, COND_WORKSPACE_SUCCEEDED, False, "TuningFailed")
            ws.status.state = "Failed"
            return ReconcileResult()
        ws.status.state = "Running"
        return ReconcileResult(requeue_after_s=10.0, done=False)

    # ------------------------------------------------------------ benchmark
    def _ingest_benchmark(self, ws: Workspace) -> None:
        """Reference parity: benchmark.go:76-223 — parse
        KAITO_BENCHMARK_RESULT/CONFIG JSON lines from pod-0 logs into
        status.performance.metrics[peakTokensPerMinute]."""
        if ws.annotations.get(ANNOTATION_DISABLE_BENCHMARK) == "true":
            return
        if ws.status.performance.get("metrics"):
            return  # write-once
        logs = self.get_pod_logs(ws.namespace, f"{ws.name}-0")
        result = config = None
        for line in logs.splitlines()[-500:]:
            if BENCHMARK_RESULT_TAG in line:
                try:
                    result = json.loads(line.split(BENCHMARK_RESULT_TAG, 1)[1]
                                        .strip(" :"))
                except json.JSONDecodeError:
                    pass
            elif BENCHMARK_CONFIG_TAG in line:
                try:
                    config = json.loads(line.split(BENCHMARK_CONFIG_TAG, 1)[1]
                                        .strip(" :"))
                except json.JSONDecodeError:
                    pass
        if result:
            ws.status.performance = {
                "metrics": [{
                    "name": "peakTokensPerMinute",
                    "value": result.get("peakTokensPerMinute",
                                        result.get("value")),
                    "unit": "tokens/min",
                    "description": "stress/high-concurrency",
                }],
                "config": config or {},
            }

    def _push_status(self, ws: Workspace) -> None:
        try:
            obj = self.client.get("Workspace", ws.namespace, ws.name)
        except NotFound:
            return
        if obj["metadata"].get("finalizers", []) != ws.finalizers:
            obj["metadata"]["finalizers"] = list(ws.finalizers)
            obj = self.client.update(obj)
        obj["status"] = {
            "state": ws.status.state,
            "targetNodeCount": ws.status.targetNodeCount,
            "workerNodes": ws.status.workerNodes,
            "conditions": [c.__dict__ for c in ws.status.conditions],
            "performance": ws.status.performance,
        }
        self.client.update_status(obj)
