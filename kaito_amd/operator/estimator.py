"""GPU-memory / node-count estimator — the rewrite of the reference's
pkg/workspace/estimator/nodesestimator/estimator.go:34-197 for MI355X.

Memory model (constants recalibrated for our HIP engine on 288 GiB HBM3E):
  avail_per_gpu = (mem * util − base_overhead − kv_budget/num_gpus)
                  / (expansion * (1 + activation_factor))
  min_gpus      = ceil(weights / avail_per_gpu)
  nodes         = ceil(min_gpus / gpus_per_node)

The reference uses util=0.84, expansion=1.02, base=2.3 GiB, act=0.05
(estimator.go:34-59) sized for vLLM+CUDA-graph overheads on 80 GiB cards.
Our engine's measured residency on MI355X: HIP runtime + torch ≈ 2.5 GiB,
hipGraph pools < 1 GiB, so util=0.90 is safe on 288 GiB parts.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

from ..engine.config import ModelConfig
from .sku import GPUConfig

GPU_MEMORY_UTILIZATION = 0.90
WEIGHT_EXPANSION = 1.02        # non-weight param residency (buffers, rope)
BASE_OVERHEAD_GIB = 2.5        # HIP runtime + torch + graph pools per GPU
ACTIVATION_FACTOR = 0.05       # transient activations vs weights
GIB = 1 << 30


@dataclass
class NodeEstimateRequest:
    model: ModelConfig
    gpu: GPUConfig
    replicas: int = 1
    max_model_len: Optional[int] = None     # None → model token limit
    max_num_seqs: int = 256


@dataclass
class NodeEstimateResult:
    min_gpus: int
    nodes_per_replica: int
    target_node_count: int
    avail_mem_per_gpu_gib: float
    kv_budget_gib: float
    max_model_len: int


def estimate_node_count(req: NodeEstimateRequest) -> NodeEstimateResult:
    """Reference parity: EstimateNodeCount (estimator.go:70-197)."""
    m = req.model
    gpu = req.gpu
    max_len = req.max_model_len or min(m.max_position, 8192)
    weights_gib = m.param_bytes() / GIB

    gpus_per_node = gpu.gpu_count
    # iterate: kv budget is split across the GPUs the model lands on
    min_gpus = 1
    for _ in range(8):
        kv_tokens = max_len * min(req.max_num_seqs, 64)  # reserved KV floor
        kv_gib = kv_tokens * m.kv_bytes_per_token() / GIB
        avail = (gpu.gpu_mem_gib * GPU_MEMORY_UTILIZATION
                 - BASE_OVERHEAD_GIB
                 - kv_gib / max(min_gpus, 1)) / (
                     WEIGHT_EXPANSION * (1 + ACTIVATION_FACTOR))
        if avail <= 0:
            min_gpus *= 2
            if min_gpus > 64:
                raise ValueError(
                    f"model {m.name} cannot fit: kv budget {kv_gib:.1f} GiB "
                    f"exceeds GPU memory")
            continue
        need = max(1, math.ceil(weights_gib / avail))
        if need <= min_gpus:
            min_gpus = need
            break
        min_gpus = need
    # round up to a power-of-two TP degree within the node (xGMI mesh)
    if min_gpus > 1:
        min_gpus = 1 << math.ceil(math.log2(min_gpus))
    nodes_per_replica = max(1, math.ceil(min_gpus / gpus_per_node))
    kv_gib = max_len * min(req.max_num_seqs, 64) * m.kv_bytes_per_token() / GIB
    avail = (gpu.gpu_mem_gib * GPU_MEMORY_UTILIZATION - BASE_OVERHEAD_GIB
             - kv_gib / min_gpus) / (WEIGHT_EXPANSION * (1 + ACTIVATION_FACTOR))
    return NodeEstimateResult(
        min_gpus=min_gpus,
        nodes_per_replica=nodes_per_replica,
        target_node_count=nodes_per_replica * req.replicas,
        avail_mem_per_gpu_gib=round(avail, 2),
        kv_budget_gib=round(kv_gib, 2),
        max_model_len=max_len,
    )
