"""Embedding providers for the RAG engine.

Reference parity: presets/ragengine/embedding/huggingface_local_embedding.py
(HuggingFaceEmbedding, bge-small default) and remote_embedding.py. On
MI355X the local model's GEMMs run on MFMA through torch-ROCm/hipBLASLt;
when model weights are unavailable (air-gapped test/bench environments)
a deterministic feature-hash embedder stands in.
"""
from __future__ import annotations

import hashlib
import re
from typing import List, Optional

import numpy as np


class BaseEmbedding:
    dim: int = 384

    def embed(self, texts: List[str]) -> np.ndarray:
        raise NotImplementedError

    def embed_query(self, text: str) -> np.ndarray:
        return self.embed([text])[0]


class HashEmbedding(BaseEmbedding):
    """Deterministic feature-hash embedding (L2-normalised bag of hashed
    token n-grams). No model weights needed; used for tests and air-gapped
    deployments. Not semantically meaningful, but stable + fast."""

    def __init__(self, dim: int = 384):
        self.dim = dim

    def _tokens(self, text: str) -> List[str]:
        toks = re.findall(r"\w+", text.lower())
        return toks + [" ".join(p) for p in zip(toks, toks[1:])]

    def embed(self, texts: List[str]) -> np.ndarray:
        out = np.zeros((len(texts), self.dim), dtype=np.float32)
        for i, t in enumerate(texts):
            for tok in self._tokens(t):
                h = int.from_bytes(
                    hashlib.blake2b(tok.encode(), digest_size=8).digest(), "little")
                idx = h % self.dim
                sign = 1.0 if (h >> 63) & 1 else -1.0
                out[i, idx] += sign
            n = np.linalg.norm(out[i])
            if n > 0:
                out[i] /= n
        return out


class LocalEmbedding(BaseEmbedding):
    """HF transformer embedding (bge-small class) on ROCm: mean-pooled last
    hidden state, L2-normalised. GEMMs → hipBLASLt MFMA."""

    def __init__(self, model_id: str, device: Optional[str] = None):
        import torch
        from transformers import AutoModel, AutoTokenizer
        self.torch = torch
        self.device = device or ("cuda" if torch.cuda.is_available() else "cpu")
        self.tokenizer = AutoTokenizer.from_pretrained(model_id)
        self.model = AutoModel.from_pretrained(model_id).to(self.device).eval()
        self.dim = self.model.config.hidden_size

    def embed(self, texts: List[str]) -> np.ndarray:
        torch = self.torch
        with torch.no_grad():
            enc = self.tokenizer(texts, padding=True, truncation=True,
                                 max_length=512, return_tensors="pt").to(self.device)
            out = self.model(**enc).last_hidden_state
            mask = enc["attention_mask"].unsqueeze(-1).float()
            emb = (out * mask).sum(1) / mask.sum(1).clamp(min=1)
            emb = torch.nn.functional.normalize(emb, dim=-1)
            return emb.float().cpu().numpy()


class RemoteEmbedding(BaseEmbedding):
    """OpenAI-style /v1/embeddings endpoint."""

    def __init__(self, url: str, access_secret: str = "", dim: int = 384):
        self.url = url
        self.secret = access_secret
        self.dim = dim

    def embed(self, texts: List[str]) -> np.ndarray:
        import httpx
        headers = {"Authorization": f"Bearer {self.secret}"} if self.secret else {}
        r = httpx.post(self.url, json={"input": texts}, headers=headers,
                       timeout=60)
        r.raise_for_status()
        data = r.json()["data"]
        arr = np.array([d["embedding"] for d in data], dtype=np.float32)
        self.dim = arr.shape[1]
        return arr


def make_embedding(cfg) -> BaseEmbedding:
    if cfg.embedding_source_type == "remote" and cfg.remote_embedding_url:
        return RemoteEmbedding(cfg.remote_embedding_url,
                               cfg.remote_embedding_access_secret)
    try:
        return LocalEmbedding(cfg.local_embedding_model_id)
    except Exception:
        # air-gapped: model weights unreachable → deterministic fallback
        return HashEmbedding()
