"""Preset metadata generator — parity with the reference's offline tool
(presets/workspace/generator/generator.go: FetchModelMetadata :389,
calculateKVCacheTokenSize :660, calculateStorageSize :653; formulas also
in pkg/model/interface.go:209-213).

Computes the estimator inputs from a model's config.json (local file —
air-gapped; the reference fetches from HF Hub).
"""
from __future__ import annotations

import json
import math
import os
from dataclasses import dataclass
from typing import Optional

DTYPE_BYTES = {"bfloat16": 2, "float16": 2, "float32": 4}


@dataclass
class PresetMetadata:
    name: str
    hidden_size: int
    num_layers: int
    num_kv_heads: int
    head_dim: int
    bytes_per_token: int          # KV bytes/token (2*layers*kvh*hd*dtype)
    total_param_bytes: int
    disk_storage_gib: int         # size*2.5 + 48, rounded up to 10
    model_token_limit: int


def kv_bytes_per_token(num_layers: int, kv_heads: int, head_dim: int,
                       dtype_bytes: int = 2) -> int:
    """Reference parity: BytesPerToken = 2*layers*kvHeads*headDim*dtype."""
    return 2 * num_layers * kv_heads * head_dim * dtype_bytes


def disk_storage_gib(param_bytes: int) -> int:
    """Reference parity: size*2.5 + 48 GiB, rounded up to 10
    (interface.go:209-213)."""
    gib = param_bytes / (1 << 30)
    raw = gib * 2.5 + 48
    return int(math.ceil(raw / 10.0) * 10)


def generate_preset_metadata(config_path: str,
                             name: Optional[str] = None,
                             param_bytes: Optional[int] = None
                             ) -> PresetMetadata:
    with open(config_path) as f:
        cfg = json.load(f)
    hidden = cfg["hidden_size"]
    layers = cfg.get("num_hidden_layers", cfg.get("num_layers"))
    heads = cfg.get("num_attention_heads")
    kvh = cfg.get("num_key_value_heads", heads)
    head_dim = cfg.get("head_dim", hidden // heads)
    dtype = DTYPE_BYTES.get(cfg.get("torch_dtype", "bfloat16"), 2)
    limit = cfg.get("max_position_embeddings", 8192)
    if param_bytes is None:
        inter = cfg.get("intermediate_size", 4 * hidden)
        vocab = cfg.get("vocab_size", 32000)
        qkv = hidden * (heads + 2 * kvh) * head_dim
        per_layer = qkv + heads * head_dim * hidden + 3 * hidden * inter
        tie = cfg.get("tie_word_embeddings", False)
        param_bytes = dtype * (layers * per_layer +
                               vocab * hidden * (1 if tie else 2))
    return PresetMetadata(
        name=name or os.path.basename(os.path.dirname(config_path)) or "model",
        hidden_size=hidden, num_layers=layers, num_kv_heads=kvh,
        head_dim=head_dim,
        bytes_per_token=kv_bytes_per_token(layers, kvh, head_dim, dtype),
        total_param_bytes=param_bytes,
        disk_storage_gib=disk_storage_gib(param_bytes),
        model_token_limit=limit)


def catalog_row(model_dir: str, name: str = "",
                runtime: str = "") -> dict:
    """One model-catalog row (reference: model_catalog.yaml entries), from
    a local model directory. Prefers the safetensors index's total_size
    over the parameter-count estimate."""
    from ..engine.config import ModelConfig
    cfg_path = os.path.join(model_dir, "config.json")
    with open(cfg_path) as f:
        hf = json.load(f)
    mc = ModelConfig.from_hf_config(hf, name=name)
    size = None
    idx = os.path.join(model_dir, "model.safetensors.index.json")
    if os.path.exists(idx):
        with open(idx) as f:
            size = json.load(f).get("metadata", {}).get("total_size")
    md = generate_preset_metadata(cfg_path, name=name or mc.name,
                                  param_bytes=size)
    return {
        "name": md.name,
        "runtime": runtime or mc.runtime,
        "architecture": (hf.get("architectures") or ["unknown"])[0],
        "hiddenSize": md.hidden_size,
        "numLayers": md.num_layers,
        "numKeyValueHeads": md.num_kv_heads,
        "headDim": md.head_dim,
        "bytesPerToken": md.bytes_per_token,
        "modelTokenLimit": md.model_token_limit,
        "totalFileSizeBytes": md.total_param_bytes,
        "diskStorageRequirementGiB": md.disk_storage_gib,
    }


def main(argv=None):
    """CLI: python -m kaito_amd.utils.preset_generator <model_dir>
    [--name N] [--runtime native|transformers] [--append-to catalog.json]"""
    import argparse
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("model_dir")
    p.add_argument("--name", default="")
    p.add_argument("--runtime", default="",
                   choices=["", "native", "transformers"])
    p.add_argument("--append-to", default=None)
    args = p.parse_args(argv)
    row = catalog_row(args.model_dir, args.name, args.runtime)
    print(json.dumps(row, indent=2))
    if args.append_to:
        cat = []
        if os.path.exists(args.append_to):
            with open(args.append_to) as f:
                cat = json.load(f)
        cat = [r for r in cat if r["name"] != row["name"]] + [row]
        with open(args.append_to, "w") as f:
            json.dump(cat, f, indent=2)


if __name__ == "__main__":
    main()
