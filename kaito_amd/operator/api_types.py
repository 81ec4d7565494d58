"""Workspace / InferenceSet / RAGEngine API types + admission validation.

Python re-implementation of the reference's CRD Go structs
(api/v1beta1/workspace_types.go:298-306, inferenceset_types.go,
api/v1alpha1/ragengine_types.go:86-112) keeping field names, condition
types and label keys byte-compatible (SURVEY.md §8 contract appendix) so
ecosystem tooling can diff/consume unchanged.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

# ---- labels / annotations (api/v1beta1/labels.go) --------------------------
LABEL_WORKSPACE_NAME = "kaito.sh/workspace"
LABEL_WORKSPACE_NAMESPACE = "kaito.sh/workspacenamespace"
ANNOTATION_RUNTIME = "kaito.sh/runtime"
ANNOTATION_ENABLE_LB = "kaito.sh/enablelb"
ANNOTATION_BYPASS_RESOURCE_CHECKS = "kaito.sh/bypass-resource-checks"
ANNOTATION_DISABLE_BENCHMARK = "kaito.sh/disable-benchmark"
ANNOTATION_PERFORMANCE_MODE = "kaito.sh/performance-mode"
ANNOTATION_USE_LOCAL_WEIGHTS = "kaito.sh/use-local-weights"
LABEL_INFERENCE_ROLE = "kaito.sh/inference-role"
LABEL_UPGRADE_TO_VERSION = "kaito.sh/upgrade-to-version"
LABEL_INFERENCESET_CREATED_BY = "inferenceset.kaito.sh/created-by"
LABEL_MRI_CREATED_BY = "multiroleinference.kaito.sh/created-by"
ANNOTATION_WORKSPACE_REVISION = "workspace.kaito.io/revision"
ANNOTATION_INFERENCESET_REVISION = "inferenceset.kaito.io/revision"
ANNOTATION_RAGENGINE_REVISION = "ragengine.kaito.io/revision"
ANNOTATION_WORKSPACE_HASH = "workspace.kaito.io/hash"
ANNOTATION_INFERENCESET_HASH = "inferenceset.kaito.io/hash"
FINALIZER_WORKSPACE = "workspace.finalizer.kaito.sh"
FINALIZER_INFERENCESET = "inferenceset.finalizer.kaito.sh"

# ---- condition types (api/v1beta1/condition_types.go:20-73) ----------------
COND_NODECLAIM_READY = "NodeClaimReady"
COND_NODES_READY = "NodesReady"
COND_RESOURCE_READY = "ResourceReady"
COND_INFERENCE_READY = "InferenceReady"
COND_JOB_STARTED = "JobStarted"
COND_WORKSPACE_SUCCEEDED = "WorkspaceSucceeded"
COND_WORKSPACE_DELETING = "WorkspaceDeleting"
COND_BENCHMARK_COMPLETED = "BenchmarkCompleted"
COND_MODELMIRROR_READY = "ModelMirrorReady"
COND_INFERENCESET_READY = "InferenceSetReady"
COND_INFERENCESET_DELETING = "InferenceSetDeleting"
COND_SCALING_DOWN_COMPLETED = "ScalingDownCompleted"
COND_RAGENGINE_SERVICE_READY = "ServiceReady"
COND_RAGENGINE_SUCCEEDED = "RAGEngineSucceeded"
COND_RAGENGINE_DELETING = "RAGEngineDeleting"

WORKSPACE_STATES = ("Pending", "Ready", "NotReady", "Running", "Succeeded",
                    "Failed")


class ValidationError(ValueError):
    pass


@dataclass
class Condition:
    type: str
    status: str                       # "True" | "False" | "Unknown"
    reason: str = ""
    message: str = ""
    lastTransitionTime: Optional[str] = None


@dataclass
class PartitionSpec:
    """workspace_types.go:72-90 (MIG partitioning on NVIDIA; on MI355X
    partitioning maps to SR-IOV/CPX partitions — validated but unexpanded)."""
    partitionType: Optional[str] = None
    partitionCount: Optional[int] = None


@dataclass
class ResourceSpec:
    instanceType: str = ""
    labelSelector: Dict[str, Any] = field(default_factory=dict)
    preferredNodes: List[str] = field(default_factory=list)
    count: Optional[int] = None
    partition: Optional[PartitionSpec] = None


@dataclass
class AdapterSpec:
    source: Dict[str, Any] = field(default_factory=dict)   # {name, image, ...}
    strength: Optional[str] = None


@dataclass
class PresetSpec:
    name: str = ""
    presetOptions: Dict[str, Any] = field(default_factory=dict)
    accessMode: str = "public"


@dataclass
class InferenceSpec:
    preset: Optional[PresetSpec] = None
    template: Optional[Dict[str, Any]] = None     # raw podTemplate
    config: str = ""                              # ConfigMap name
    adapters: List[AdapterSpec] = field(default_factory=list)


@dataclass
class DataSource:
    name: str = ""
    urls: List[str] = field(default_factory=list)
    image: str = ""
    volumeSource: Optional[Dict[str, Any]] = None


@dataclass
class DataDestination:
    image: str = ""
    imagePushSecret: str = ""
    volumeSource: Optional[Dict[str, Any]] = None


@dataclass
class TuningSpec:
    preset: Optional[PresetSpec] = None
    method: str = "lora"                          # lora | qlora
    config: str = ""
    input: Optional[DataSource] = None
    output: Optional[DataDestination] = None


@dataclass
class WorkspaceStatus:
    conditions: List[Condition] = field(default_factory=list)
    workerNodes: List[str] = field(default_factory=list)
    targetNodeCount: int = 0
    state: str = "Pending"
    performance: Dict[str, Any] = field(default_factory=dict)


@dataclass
class Workspace:
    """api/v1beta1/workspace_types.go:298-306."""
    name: str
    namespace: str = "default"
    resource: ResourceSpec = field(default_factory=ResourceSpec)
    inference: Optional[InferenceSpec] = None
    tuning: Optional[TuningSpec] = None
    annotations: Dict[str, str] = field(default_factory=dict)
    labels: Dict[str, str] = field(default_factory=dict)
    status: WorkspaceStatus = field(default_factory=WorkspaceStatus)
    # set when the API server marks the object for deletion
    deletionTimestamp: Optional[str] = None
    finalizers: List[str] = field(default_factory=list)

    def validate(self, sku_handler=None, known_presets=None) -> None:
        """Admission validation — the semantic checks from
        api/v1beta1/workspace_validation.go (974 L), condensed."""
        if (self.inference is None) == (self.tuning is None):
            raise ValidationError(
                "exactly one of inference or tuning must be set")
        if not self.resource.instanceType and not self.resource.labelSelector:
            raise ValidationError(
                "resource.instanceType or labelSelector required")
        if sku_handler is not None and self.resource.instanceType:
            if sku_handler.get_gpu_config(self.resource.instanceType) is None \
                    and ANNOTATION_BYPASS_RESOURCE_CHECKS not in self.annotations:
                raise ValidationError(
                    f"unsupported instanceType {self.resource.instanceType!r}")
        if self.inference is not None:
            if self.inference.preset is None and self.inference.template is None:
                raise ValidationError("inference needs preset or template")
            if self.inference.preset and known_presets is not None and \
                    self.inference.preset.name not in known_presets:
                raise ValidationError(
                    f"unknown preset {self.inference.preset.name!r}")
            if len(self.inference.adapters) > 10:
                raise ValidationError("at most 10 adapters supported")
            names = [a.source.get("name") for a in self.inference.adapters]
            if len(names) != len(set(names)):
                raise ValidationError("adapter names must be unique")
        if self.tuning is not None:
            if self.tuning.method not in ("lora", "qlora"):
                raise ValidationError(f"unknown tuning method {self.tuning.method}")
            if self.tuning.input is None or self.tuning.output is None:
                raise ValidationError("tuning requires input and output")


@dataclass
class InferenceSetSpec:
    replicas: int = 1
    workspaceTemplate: Optional[Workspace] = None
    upgradeStrategy: str = "Surge"                # Surge | InPlace
    maintenanceWindow: str = ""                   # cron expression


@dataclass
class InferenceSetStatus:
    readyReplicas: int = 0
    replicas: int = 0
    selector: str = ""
    aggregatedPeakTokensPerMinute: float = 0.0
    conditions: List[Condition] = field(default_factory=list)


@dataclass
class InferenceSet:
    """api/v1beta1/inferenceset_types.go."""
    name: str
    namespace: str = "default"
    spec: InferenceSetSpec = field(default_factory=InferenceSetSpec)
    status: InferenceSetStatus = field(default_factory=InferenceSetStatus)
    deletionTimestamp: Optional[str] = None
    finalizers: List[str] = field(default_factory=list)

    def validate(self) -> None:
        if self.spec.replicas < 0:
            raise ValidationError("replicas must be >= 0")
        if self.spec.workspaceTemplate is None:
            raise ValidationError("workspaceTemplate required")
        if self.spec.upgradeStrategy not in ("Surge", "InPlace"):
            raise ValidationError(
                f"unknown upgradeStrategy {self.spec.upgradeStrategy}")
        self.spec.workspaceTemplate.validate()


@dataclass
class RAGEngineSpec:
    """api/v1alpha1/ragengine_types.go:86-112."""
    compute: ResourceSpec = field(default_factory=ResourceSpec)
    embedding: Dict[str, Any] = field(default_factory=dict)
    inferenceService: Dict[str, Any] = field(default_factory=dict)
    storage: Dict[str, Any] = field(default_factory=dict)
    guardrails: Dict[str, Any] = field(default_factory=dict)
    indexServiceName: str = ""
    queryServiceName: str = ""


@dataclass
class RAGEngine:
    name: str
    namespace: str = "default"
    spec: RAGEngineSpec = field(default_factory=RAGEngineSpec)
    status: Dict[str, Any] = field(default_factory=dict)
    deletionTimestamp: Optional[str] = None

    def validate(self) -> None:
        emb = self.spec.embedding
        if not emb:
            raise ValidationError("embedding spec required")
        local = emb.get("local")
        remote = emb.get("remote")
        if (local is None) == (remote is None):
            raise ValidationError(
                "exactly one of embedding.local / embedding.remote")
        if not self.spec.inferenceService.get("url"):
            raise ValidationError("inferenceService.url required")
