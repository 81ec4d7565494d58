"""Tokenizer abstraction: HF tokenizer when weights/tokenizer files exist,
byte-fallback tokenizer for synthetic/random-weight serving (no network)."""
from __future__ import annotations

from typing import List, Optional


class ByteTokenizer:
    """Reversible UTF-8 byte tokenizer: id = byte + 3 (0=pad,1=bos,2=eos).
    Used for synthetic serving where model weights are random anyway."""

    bos_token_id = 1
    eos_token_id = 2

    def __init__(self, vocab_size: int = 128256):
        self.vocab_size = vocab_size

    def encode(self, text: str) -> List[int]:
        return [self.bos_token_id] + [b + 3 for b in text.encode("utf-8")]

    def decode(self, ids: List[int]) -> str:
        bs = bytes(max(0, min(255, i - 3)) for i in ids if i > 2)
        return bs.decode("utf-8", errors="replace")

    def apply_chat_template(self, messages, add_generation_prompt=True,
                            tokenize=False):
        text = "".join(f"<{m['role']}>{m['content']}</{m['role']}>"
                       for m in messages)
        if add_generation_prompt:
            text += "<assistant>"
        return text


def load_tokenizer(path: Optional[str], vocab_size: int = 128256):
    if path:
        try:
            from transformers import AutoTokenizer
            return AutoTokenizer.from_pretrained(path)
        except Exception:
            pass
    return ByteTokenizer(vocab_size)
