"""Parallelism auto-configuration + engine command builder.

Rewrite of the reference's 3-tier planner (pkg/model/interface.go:543-573)
and buildVLLMInferenceCommand (:394-528), retargeted at an 8-GPU xGMI node:

  tier 1 (DP):  model < 50% of ONE 288 GiB MI355X → data-parallel engine
                replicas, tp=1 (most <70B models on MI355X — a bigger DP
                tier than on 80 GiB NVIDIA parts)
  tier 2 (TP):  model fits the node → tensor-parallel over xGMI
  tier 3 (PP):  multi-node → pipeline stages across nodes, TP within node;
                torchrun-style TCP rendezvous keyed on POD_INDEX + headless
                service replaces the reference's Ray bootstrap
                (interface.go:577-601).
"""
from __future__ import annotations

import shlex
from dataclasses import dataclass
from typing import List, Optional

from ..engine.config import ModelConfig
from .estimator import (GIB, GPU_MEMORY_UTILIZATION, BASE_OVERHEAD_GIB)
from .sku import GPUConfig


@dataclass
class ParallelPlan:
    data_parallel: int = 1
    tensor_parallel: int = 1
    pipeline_parallel: int = 1
    num_nodes: int = 1
    gpus_per_node: int = 1
    kv_offload: bool = True     # CPU KV offload (off for tier-1 small models)

    @property
    def world_size(self) -> int:
        return self.data_parallel * self.tensor_parallel * self.pipeline_parallel


def configure_parallelism(model: ModelConfig, gpu: GPUConfig,
                          num_nodes: int = 1) -> ParallelPlan:
    """Reference parity: configureParallelism (interface.go:543-573)."""
    weights_gib = model.param_bytes() / GIB
    per_gpu_budget = gpu.gpu_mem_gib * GPU_MEMORY_UTILIZATION - BASE_OVERHEAD_GIB
    gpus = gpu.gpu_count

    if num_nodes <= 1:
        if weights_gib < 0.5 * per_gpu_budget:
            # tier 1: replicate — serve with DP engine replicas
            return ParallelPlan(data_parallel=gpus, tensor_parallel=1,
                                num_nodes=1, gpus_per_node=gpus,
                                kv_offload=False)
        # tier 2: shard across the xGMI mesh. MLA models never offload:
        # the engine keeps the compressed latent cache on-device (its
        # aliased (c, c) pair would double offloaded bytes — engine.py)
        return ParallelPlan(tensor_parallel=gpus, num_nodes=1,
                            gpus_per_node=gpus,
                            kv_offload=not model.is_mla)
    # tier 3: pipeline across nodes, TP within each node
    return ParallelPlan(tensor_parallel=gpus, pipeline_parallel=num_nodes,
                        num_nodes=num_nodes, gpus_per_node=gpus,
                        kv_offload=not model.is_mla)


def build_inference_command(model: ModelConfig, gpu: GPUConfig,
                            plan: Optional[ParallelPlan] = None,
                            max_model_len: Optional[int] = None,
                            config_file: Optional[str] = None,
                            weights_path: Optional[str] = None,
                            enable_lora: bool = False,
                            port: int = 5000) -> List[str]:
    """Builds the pod container command — analog of GetInferenceCommand
    (interface.go:345, buildVLLMInferenceCommand :394-528) targeting our
    kaito_amd.server.entrypoint instead of vLLM's api server."""
    plan = plan or configure_parallelism(model, gpu)
    cmd = ["python3", "-m", "kaito_amd.server.entrypoint",
           "--model", model.name,
           "--port", str(port),
           "--tensor-parallel-size", str(plan.tensor_parallel)]
    if plan.data_parallel > 1:
        cmd += ["--data-parallel-size", str(plan.data_parallel)]
    if plan.pipeline_parallel > 1:
        cmd += ["--pipeline-parallel-size", str(plan.pipeline_parallel)]
    cmd += ["--max-model-len", str(max_model_len) if max_model_len else "auto"]
    if weights_path:
        cmd += ["--weights-path", weights_path]
    if config_file:
        cmd += ["--kaito-config-file", config_file]
    if enable_lora:
        cmd += ["--enable-lora"]
    return cmd


def build_multinode_command(model: ModelConfig, gpu: GPUConfig,
                            plan: ParallelPlan, headless_service: str,
                            port: int = 5000) -> str:
    """Tier-3 launcher: torchrun rendezvous on the pod-0 DNS name via the
    headless Service (replaces the reference's Ray leader/worker bootstrap,
    interface.go:577-601; POD_INDEX comes from the statefulset pod index
    label, preset_inferences.go:1001-1008)."""
    inner = build_inference_command(model, gpu, plan, port=port)
    launcher = (
        "python3 -m torch.distributed.run "
        f"--nnodes={plan.num_nodes} --nproc-per-node={plan.gpus_per_node} "
        "--node-rank=${POD_INDEX} "
        f"--master-addr={headless_service} --master-port=29500 "
        + " ".join(shlex.quote(c) for c in inner[1:]))
    return launcher
