"""GPU partitioning — the MI355X analog of the reference's MIG support
(pkg/utils/mig profile table, PartitionSpec workspace_types.go:72-90,
isMIGNode helpers.go:144-149, MIG single-slice estimator check
estimator.go:179-188).

CDNA3/4 Instinct parts partition by compute (SPX = one partition, CPX =
one partition per XCD) and memory (NPS1/NPS2). On MI355X CPX yields 8
partitions of 32 CUs with 288/8 = 36 GiB HBM3E each; a partition is
exposed as its own device, so the estimator treats it as a small GPU.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

from .api_types import PartitionSpec, ValidationError
from .sku import GPUConfig


@dataclass(frozen=True)
class PartitionProfile:
    name: str                 # spx | cpx
    partitions_per_gpu: int
    cus_per_partition: int
    mem_gib_per_partition: int


# MI355X (gfx950, 256 CUs, 288 GiB, 8 XCDs)
MI355X_PROFILES: Dict[str, PartitionProfile] = {
    "spx": PartitionProfile("spx", 1, 256, 288),
    "cpx": PartitionProfile("cpx", 8, 32, 36),
}

# resource name exposed per partition (device-plugin contract)
PARTITION_RESOURCE = "amd.com/gpu-partition"


def validate_partition(spec: Optional[PartitionSpec],
                       gpu: GPUConfig) -> Optional[PartitionProfile]:
    if spec is None or spec.partitionType is None:
        return None
    prof = MI355X_PROFILES.get(spec.partitionType.lower())
    if prof is None:
        raise ValidationError(
            f"unknown partitionType {spec.partitionType!r}; "
            f"supported: {sorted(MI355X_PROFILES)}")
    if spec.partitionCount and spec.partitionCount > \
            prof.partitions_per_gpu * gpu.gpu_count:
        raise ValidationError(
            f"partitionCount {spec.partitionCount} exceeds "
            f"{prof.partitions_per_gpu * gpu.gpu_count} available "
            f"({prof.name} on {gpu.gpu_count} GPUs)")
    return prof


def partitioned_gpu_config(gpu: GPUConfig,
                           prof: PartitionProfile) -> GPUConfig:
    """GPUConfig viewed as partitions — feeds the estimator so a model must
    fit inside ONE partition (the reference's MIG single-slice rule,
    estimator.go:179-188)."""
    return GPUConfig(
        sku=f"{gpu.sku}/{prof.name}",
        gpu_count=gpu.gpu_count * prof.partitions_per_gpu,
        gpu_mem_gib=prof.mem_gib_per_partition,
        gpu_model=gpu.gpu_model + f" ({prof.name.upper()})",
        gfx_arch=gpu.gfx_arch,
        xgmi_links=0,      # partitions do not span xGMI
        nvme_enabled=gpu.nvme_enabled)
