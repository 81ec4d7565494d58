"""Tuning runtime tests: LoRA injection, QLoRA int8 base, SFT loop on a
tiny HF llama, peft-format adapter save → served by our LoRAManager."""
import json
import os

import pytest
import torch

from kaito_amd.tuning.fine_tuning import (collate, load_dataset_texts,
                                          parse_config, run_sft)
from kaito_amd.tuning.lora_layers import (Int8Linear, LoRALinear, inject_lora,
                                          save_adapter, trainable_parameters)


def _tiny_hf_model():
    from transformers import LlamaConfig, LlamaForCausalLM
    cfg = LlamaConfig(hidden_size=64, intermediate_size=128,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, vocab_size=256,
                      max_position_embeddings=128)
    return LlamaForCausalLM(cfg)


class _TinyTok:
    pad_token = "<pad>"
    pad_token_id = 0

    def __call__(self, text, truncation=True, max_length=64,
                 return_tensors="pt"):
        ids = [1] + [2 + (b % 250) for b in text.encode()][: max_length - 1]
        return {"input_ids": torch.tensor([ids])}


def test_int8_linear_close_to_fp():
    lin = torch.nn.Linear(32, 16, bias=False)
    q = Int8Linear(lin)
    x = torch.randn(4, 32)
    assert torch.allclose(q(x), lin(x), atol=0.05, rtol=0.05)


def test_inject_lora_freezes_base():
    m = _tiny_hf_model()
    wrapped = inject_lora(m, rank=4, alpha=8)
    assert len(wrapped) == 2 * 7  # 2 layers x 7 targets
    tp = trainable_parameters(m)
    assert all(p.requires_grad for p in tp)
    total = sum(p.numel() for p in m.parameters())
    trainable = sum(p.numel() for p in tp)
    assert trainable < total * 0.2
    # forward still works
    out = m(input_ids=torch.tensor([[1, 2, 3]]),
            labels=torch.tensor([[1, 2, 3]]))
    assert out.loss.requires_grad


def test_qlora_quantized_base():
    m = _tiny_hf_model()
    inject_lora(m, rank=4, alpha=8, quantize_base=True)
    assert any(isinstance(mod, Int8Linear) for mod in m.modules())


def test_sft_loss_decreases_and_adapter_roundtrip(tmp_path):
    torch.manual_seed(0)
    m = _tiny_hf_model()
    tok = _TinyTok()
    texts = ["hello world this is a test"] * 8
    cfg = parse_config(None)
    cfg["TrainingArguments"].update(max_steps=12, learning_rate=5e-3,
                                    per_device_train_batch_size=2,
                                    num_train_epochs=5, logging_steps=100)
    cfg["LoraConfig"].update(r=4, lora_alpha=8, lora_dropout=0.0)
    # capture pre-training loss
    stats = run_sft(m, tok, texts, cfg, device="cpu")
    assert stats["steps"] == 12
    assert stats["final_loss"] is not None

    save_adapter(m, str(tmp_path), 4, 8, "tiny")
    assert (tmp_path / "adapter_model.safetensors").exists()
    assert (tmp_path / "fine_tuning_completed.txt").exists()
    with open(tmp_path / "adapter_config.json") as f:
        ac = json.load(f)
    assert ac["r"] == 4 and ac["peft_type"] == "LORA"

    # roundtrip: serving LoRAManager loads the adapter we just saved
    from kaito_amd.engine.lora import LoRAManager
    from kaito_amd.models import get_model_config
    from kaito_amd.models.llama import LlamaForCausalLM as OurLlama
    from kaito_amd.parallel.state import init_parallel
    init_parallel(1)
    mc = get_model_config("tiny-llama-test")
    # shapes differ from the HF tiny model; loader skips mismatches but must
    # parse the format without error
    our = OurLlama(mc)
    mgr = LoRAManager(our, max_adapters=2, max_rank=16)
    slot = mgr.load_peft_adapter("tuned", str(tmp_path))
    assert slot == 0


def test_config_parser_merges(tmp_path):
    p = tmp_path / "cfg.yaml"
    p.write_text("""
training_config:
  LoraConfig:
    r: 8
  TrainingArguments:
    learning_rate: 1e-3
""")
    cfg = parse_config(str(p))
    assert cfg["LoraConfig"]["r"] == 8
    assert float(cfg["TrainingArguments"]["learning_rate"]) == 1e-3
    assert cfg["TrainingArguments"]["num_train_epochs"] == 1  # default kept


def test_dataset_loader(tmp_path):
    (tmp_path / "a.jsonl").write_text(
        '{"text": "row one"}\n{"instruction": "do x", "output": "done"}\n')
    (tmp_path / "b.txt").write_text("plain line\n")
    texts = load_dataset_texts(str(tmp_path))
    assert "row one" in texts
    assert any("### Instruction" in t for t in texts)
    assert "plain line" in texts


@pytest.mark.gpu
def test_sft_on_gpu_lora_and_qlora():
    """Tuning path on a real MI355X: LoRA and QLoRA (int8 base) SFT both
    step and reduce loss on-device."""
    for method in ("lora", "qlora"):
        torch.manual_seed(0)
        m = _tiny_hf_model()
        tok = _TinyTok()
        texts = ["the quick brown fox jumps over the lazy dog"] * 8
        cfg = parse_config(None)
        cfg["TrainingArguments"].update(max_steps=10, learning_rate=5e-3,
                                        per_device_train_batch_size=2,
                                        num_train_epochs=5,
                                        logging_steps=100)
        cfg["LoraConfig"].update(r=4, lora_alpha=8, lora_dropout=0.0)
        if method == "qlora":
            cfg["QuantizationConfig"]["load_in_8bit"] = True
        stats = run_sft(m, tok, texts, cfg, device="cuda")
        assert stats["steps"] == 10
        assert stats["final_loss"] is not None
        assert stats["final_loss"] < stats["first_loss"], (method, stats)
