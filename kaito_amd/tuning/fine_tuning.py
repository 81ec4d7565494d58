"""LoRA / QLoRA supervised fine-tuning entrypoint.

Reference parity: presets/workspace/tuning/text-generation/fine_tuning.py
(YAML-config-driven peft+TRL SFT, accelerate multi-GPU, adapter-only save +
completion marker). This implementation is torch-ROCm native (no
peft/trl/bitsandbytes): HF transformers model + in-repo LoRA injection,
DDP via torch.distributed when WORLD_SIZE>1 (launched by the operator's
accelerate-style command, pkg/model/interface.go:722-731 analog).

Config YAML keys mirror the reference's parser (training_config:
ModelConfig / QuantizationConfig / LoraConfig / TrainingArguments /
DatasetConfig / DataCollator).
"""
from __future__ import annotations

import argparse
import json
import logging
import os
import time
from typing import Dict, List, Optional

import torch
import yaml

from .lora_layers import inject_lora, save_adapter, trainable_parameters

logger = logging.getLogger("kaito_amd.tuning")


def parse_config(path: Optional[str]) -> Dict:
    cfg = {
        "ModelConfig": {"pretrained_model_name_or_path": None,
                        "torch_dtype": "bfloat16"},
        "QuantizationConfig": {"load_in_8bit": False, "load_in_4bit": False},
        "LoraConfig": {"r": 16, "lora_alpha": 32, "lora_dropout": 0.05},
        "TrainingArguments": {"num_train_epochs": 1, "per_device_train_batch_size": 1,
                              "learning_rate": 2e-4, "max_steps": -1,
                              "gradient_accumulation_steps": 1,
                              "output_dir": "/mnt/results",
                              "save_steps": 0, "logging_steps": 10,
                              "max_seq_length": 512},
        "DatasetConfig": {"shuffle_dataset": True, "train_test_split": 1.0,
                          "context_column": "text"},
    }
    if path and os.path.exists(path):
        with open(path) as f:
            user = yaml.safe_load(f) or {}
        user = user.get("training_config", user)
        for k, v in user.items():
            if k in cfg and isinstance(v, dict):
                cfg[k].update(v)
            else:
                cfg[k] = v
    return cfg


def load_dataset_texts(data_dir: str, column: str = "text") -> List[str]:
    """Load training texts from /mnt/data (json/jsonl/csv/txt files)."""
    import glob
    texts: List[str] = []
    for path in sorted(glob.glob(os.path.join(data_dir, "**/*"), recursive=True)):
        if os.path.isdir(path):
            continue
        try:
            if path.endswith((".json", ".jsonl", ".dat")):
                with open(path) as f:
                    content = f.read().strip()
                rows = []
                try:
                    data = json.loads(content)
                    rows = data if isinstance(data, list) else [data]
                except json.JSONDecodeError:
                    rows = [json.loads(ln) for ln in content.splitlines() if ln.strip()]
                for r in rows:
                    if isinstance(r, dict):
                        if column in r:
                            texts.append(str(r[column]))
                        elif {"instruction", "output"} <= set(r):
                            texts.append(f"### Instruction:\n{r['instruction']}"
                                         f"\n### Response:\n{r['output']}")
                    else:
                        texts.append(str(r))
            elif path.endswith((".txt", ".csv")):
                with open(path) as f:
                    texts.extend(ln.strip() for ln in f if ln.strip())
        except Exception as e:  # noqa: BLE001
            logger.warning("skipping %s: %s", path, e)
    return texts


class TextDataset(torch.utils.data.Dataset):
    def __init__(self, texts, tokenizer, max_len):
        self.enc = [tokenizer(t, truncation=True, max_length=max_len,
                              return_tensors="pt") for t in texts]

    def __len__(self):
        return len(self.enc)

    def __getitem__(self, i):
        ids = self.enc[i]["input_ids"][0]
        return {"input_ids": ids, "labels": ids.clone()}


def collate(batch, pad_id=0):
    ml = max(len(b["input_ids"]) for b in batch)
    ids = torch.full((len(batch), ml), pad_id, dtype=torch.long)
    labels = torch.full((len(batch), ml), -100, dtype=torch.long)
    for i, b in enumerate(batch):
        n = len(b["input_ids"])
        ids[i, :n] = b["input_ids"]
        labels[i, :n] = b["labels"]
    return {"input_ids": ids, "labels": labels}


def run_sft(model, tokenizer, texts: List[str], cfg: Dict,
            device: str = "cpu") -> Dict:
    ta = cfg["TrainingArguments"]
    lc = cfg["LoraConfig"]
    qc = cfg["QuantizationConfig"]
    wrapped = inject_lora(model, rank=lc["r"], alpha=lc["lora_alpha"],
                          dropout=lc.get("lora_dropout", 0.0),
                          quantize_base=qc.get("load_in_8bit", False)
                          or qc.get("load_in_4bit", False))
    logger.info("LoRA-wrapped %d modules", len(wrapped))
    model.to(device).train()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world > 1:
        import torch.distributed as dist
        if not dist.is_initialized():
            dist.init_process_group(
                "nccl" if device.startswith("cuda") else "gloo")
        model = torch.nn.parallel.DistributedDataParallel(model)

    ds = TextDataset(texts, tokenizer, ta.get("max_seq_length", 512))
    sampler = (torch.utils.data.distributed.DistributedSampler(ds)
               if world > 1 else None)
    dl = torch.utils.data.DataLoader(
        ds, batch_size=ta["per_device_train_batch_size"],
        shuffle=(sampler is None and cfg["DatasetConfig"]["shuffle_dataset"]),
        sampler=sampler,
        collate_fn=lambda b: collate(b, tokenizer.pad_token_id or 0))
    opt = torch.optim.AdamW(trainable_parameters(model),
                            lr=float(ta["learning_rate"]))
    max_steps = ta.get("max_steps", -1)
    accum = max(ta.get("gradient_accumulation_steps", 1), 1)
    step = 0
    losses = []
    t0 = time.monotonic()
    for epoch in range(int(ta["num_train_epochs"])):
        for i, batch in enumerate(dl):
            batch = {k: v.to(device) for k, v in batch.items()}
            out = model(**batch)
            loss = out.loss / accum
            loss.backward()
            if (i + 1) % accum == 0:
                opt.step()
                opt.zero_grad(set_to_none=True)
                step += 1
                losses.append(float(loss) * accum)
                if step % ta.get("logging_steps", 10) == 0 and rank == 0:
                    logger.info("step %d loss %.4f", step, losses[-1])
                if device.startswith("cuda"):
                    torch.cuda.empty_cache()  # reference per-step callback
                if 0 < max_steps <= step:
                    break
        if 0 < max_steps <= step:
            break
    return {"steps": step, "final_loss": losses[-1] if losses else None,
            "first_loss": losses[0] if losses else None,
            "train_seconds": time.monotonic() - t0}


def main(argv=None):
    logging.basicConfig(level=logging.INFO)
    p = argparse.ArgumentParser()
    p.add_argument("--model", default=os.environ.get("KAITO_MODEL", ""))
    p.add_argument("--weights-path", default=os.environ.get("KAITO_WEIGHTS_PATH"))
    p.add_argument("--method", default="lora", choices=["lora", "qlora"])
    p.add_argument("--config", default="/mnt/config/training_config.yaml")
    p.add_argument("--data-dir", default="/mnt/data")
    p.add_argument("--output-dir", default=None)
    p.add_argument("--num-processes", type=int, default=1)
    args = p.parse_args(argv)

    cfg = parse_config(args.config)
    if args.method == "qlora":
        cfg["QuantizationConfig"]["load_in_4bit"] = True
    out_dir = args.output_dir or cfg["TrainingArguments"]["output_dir"]

    from transformers import AutoModelForCausalLM, AutoTokenizer
    src = args.weights_path or \
        cfg["ModelConfig"].get("pretrained_model_name_or_path")
    model = AutoModelForCausalLM.from_pretrained(
        src, torch_dtype=torch.bfloat16 if torch.cuda.is_available()
        else torch.float32)
    tokenizer = AutoTokenizer.from_pretrained(src)
    if tokenizer.pad_token is None:
        tokenizer.pad_token = tokenizer.eos_token

    texts = load_dataset_texts(args.data_dir,
                               cfg["DatasetConfig"]["context_column"])
    if not texts:
        raise SystemExit(f"no training data found under {args.data_dir}")
    device = "cuda" if torch.cuda.is_available() else "cpu"
    stats = run_sft(model, tokenizer, texts, cfg, device)
    if int(os.environ.get("RANK", "0")) == 0:
        lc = cfg["LoraConfig"]
        target = model.module if hasattr(model, "module") else model
        save_adapter(target, out_dir, lc["r"], lc["lora_alpha"], src or "")
        print(json.dumps({"status": "completed", **stats}))


if __name__ == "__main__":
    main()
