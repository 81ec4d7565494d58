"""MI355X SKU tables — the rewrite of the reference's pkg/sku
(cloud_sku_handler.go:25-49, azure_sku_handler.go:18-49).

The reference's tables are NVIDIA (A10/A100/H100/H200 over NVLink); this is
the MI355X-native equivalent: 288 GiB HBM3E per GPU, gfx950, xGMI mesh
(7 point-to-point links per GPU on an 8-GPU node). BYO nodes are resolved
from amd.com/* node labels (reference: GetGPUConfigFromNodeLabels,
pkg/sku/helpers.go:75-119 for nvidia.com/*).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional


@dataclass(frozen=True)
class GPUConfig:
    sku: str
    gpu_count: int
    gpu_mem_gib: int          # per GPU
    gpu_model: str
    gfx_arch: str = "gfx950"
    xgmi_links: int = 0       # p2p links per GPU (0 = single GPU / PCIe)
    nvme_enabled: bool = False

    @property
    def total_gpu_mem_gib(self) -> int:
        return self.gpu_count * self.gpu_mem_gib

    def supports_bfloat16(self) -> bool:
        # all CDNA2+ (gfx90a/gfx942/gfx950) support bf16 MFMA
        return self.gfx_arch >= "gfx90a"

    def scale_to_count(self, count: int) -> "GPUConfig":
        """Reference parity: ScaleGPUConfigToCount (pkg/sku/helpers.go:123)."""
        return GPUConfig(self.sku, count, self.gpu_mem_gib, self.gpu_model,
                         self.gfx_arch, self.xgmi_links, self.nvme_enabled)


MI355X = "AMD Instinct MI355X"
MI300X = "AMD Instinct MI300X"

# Azure-style instance names for MI35x-class nodes (ND MI300X v5 is the
# published Azure shape; MI355X entries follow the same naming scheme).
AZURE_SKUS: List[GPUConfig] = [
    GPUConfig("Standard_ND96isr_MI355X_v1", 8, 288, MI355X, "gfx950", 7, True),
    GPUConfig("Standard_ND48isr_MI355X_v1", 4, 288, MI355X, "gfx950", 3, True),
    GPUConfig("Standard_ND24isr_MI355X_v1", 2, 288, MI355X, "gfx950", 1, True),
    GPUConfig("Standard_NC12s_MI355X_v1", 1, 288, MI355X, "gfx950", 0, False),
    GPUConfig("Standard_ND96isr_MI300X_v5", 8, 192, MI300X, "gfx942", 7, True),
]

AWS_SKUS: List[GPUConfig] = [
    GPUConfig("mi355x.metal-48xl", 8, 288, MI355X, "gfx950", 7, True),
]


class CloudSKUHandler:
    """Reference parity: pkg/sku/cloud_sku_handler.go:25-28."""

    def __init__(self, skus: List[GPUConfig]):
        self._by_name: Dict[str, GPUConfig] = {s.sku: s for s in skus}

    def get_gpu_configs(self) -> List[GPUConfig]:
        return list(self._by_name.values())

    def get_gpu_config(self, instance_type: str) -> Optional[GPUConfig]:
        return self._by_name.get(instance_type)

    def get_supported_skus(self) -> List[str]:
        return sorted(self._by_name)


_HANDLERS = {
    "azure": CloudSKUHandler(AZURE_SKUS),
    "aws": CloudSKUHandler(AWS_SKUS),
}


def get_sku_handler(cloud: str = "azure") -> CloudSKUHandler:
    try:
        return _HANDLERS[cloud.lower()]
    except KeyError:
        raise ValueError(f"unsupported cloud provider {cloud!r}; "
                         f"known: {sorted(_HANDLERS)}") from None


# ---- BYO nodes: amd.com/* label schema --------------------------------
LABEL_GPU_COUNT = "amd.com/gpu.count"
LABEL_GPU_MEM = "amd.com/gpu.vram"          # e.g. "288G"
LABEL_GPU_PRODUCT = "amd.com/gpu.product"   # e.g. "AMD-Instinct-MI355X"
LABEL_GFX_ARCH = "amd.com/gpu.family"       # e.g. "gfx950"


def gpu_config_from_node_labels(labels: Dict[str, str],
                                instance_type: str = "byo") -> Optional[GPUConfig]:
    """Reference parity: GetGPUConfigFromNodeLabels (pkg/sku/helpers.go:75)."""
    if LABEL_GPU_COUNT not in labels:
        return None
    count = int(labels[LABEL_GPU_COUNT])
    mem_s = labels.get(LABEL_GPU_MEM, "288G").upper().rstrip("IB").rstrip("G")
    mem = int(mem_s) if mem_s.isdigit() else 288
    product = labels.get(LABEL_GPU_PRODUCT, MI355X).replace("-", " ")
    arch = labels.get(LABEL_GFX_ARCH, "gfx950")
    links = 7 if count == 8 else max(count - 1, 0)
    return GPUConfig(instance_type, count, mem, product, arch, links)
