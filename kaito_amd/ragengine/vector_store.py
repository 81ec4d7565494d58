"""Vector store + hybrid retrieval.

Reference parity: presets/ragengine/vector_store/base.py (992 L — rwlock'd
index mutation, doc_id = sha256(text), persist/load) with the FAISS flat
backend (faiss_store.py: IndexFlatL2 in an IndexIDMap) replaced by an
MI355X-native flat index: distances via an MFMA GEMM (hipBLASLt through
torch) + the in-tree HIP top-k selection kernel (ops/csrc/topk.hip); CPU
fallback is numpy.

Hybrid retrieval (hybrid_retriever.py:60-237): vector candidates (top k*3)
+ BM25, fused 0.7·vec_norm + 0.3·1/(1+rank) — the reference's weighted
fusion (its README says RRF; the implementation is weighted fusion — we
keep the implementation semantics).
"""
from __future__ import annotations

import hashlib
import json
import os
import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np

from .bm25 import BM25Index


def doc_id_for(text: str) -> str:
    return hashlib.sha256(text.encode("utf-8")).hexdigest()


@dataclass
class Document:
    doc_id: str
    text: str
    metadata: Dict = field(default_factory=dict)


@dataclass
class RetrievedDoc:
    doc_id: str
    text: str
    score: float
    metadata: Dict = field(default_factory=dict)


class FlatIndex:
    """Flat inner-product/L2 index over normalized embeddings.

    GPU path: scores = Q @ X^T on MFMA (hipBLASLt), selection by the HIP
    top-k kernel. CPU path: numpy."""

    def __init__(self, dim: int, use_gpu: Optional[bool] = None):
        self.dim = dim
        self._vecs = np.zeros((0, dim), dtype=np.float32)
        self._ids: List[str] = []
        self._pos: Dict[str, int] = {}
        if use_gpu is None:
            try:
                import torch
                use_gpu = torch.cuda.is_available()
            except Exception:
                use_gpu = False
        self.use_gpu = use_gpu
        self._gpu_vecs = None  # lazily mirrored torch tensor

    def __len__(self):
        return len(self._ids)

    def add(self, doc_id: str, vec: np.ndarray) -> None:
        if doc_id in self._pos:
            self._vecs[self._pos[doc_id]] = vec
        else:
            self._pos[doc_id] = len(self._ids)
            self._ids.append(doc_id)
            self._vecs = np.vstack([self._vecs, vec[None, :]])
        self._gpu_vecs = None

    def remove(self, doc_id: str) -> None:
        pos = self._pos.pop(doc_id, None)
        if pos is None:
            return
        last = len(self._ids) - 1
        if pos != last:
            self._vecs[pos] = self._vecs[last]
            moved = self._ids[last]
            self._ids[pos] = moved
            self._pos[moved] = pos
        self._ids.pop()
        self._vecs = self._vecs[:last]
        self._gpu_vecs = None

    def search(self, query: np.ndarray, top_k: int):
        """Returns [(doc_id, score)] by inner product (vectors normalized →
        cosine)."""
        n = len(self._ids)
        if n == 0:
            return []
        top_k = min(top_k, n)
        if self.use_gpu:
            import torch
            from .. import ops
            if self._gpu_vecs is None:
                self._gpu_vecs = torch.from_numpy(self._vecs).to("cuda")
            q = torch.from_numpy(np.ascontiguousarray(query[None, :])).to("cuda")
            scores = (q @ self._gpu_vecs.T).float().contiguous()
            k = min(top_k, 32)
            vals = torch.empty(1, k, dtype=torch.float32, device="cuda")
            idx = torch.empty(1, k, dtype=torch.int32, device="cuda")
            ops.load_extension()
            import torch as _t
            _t.ops.kaito.topk(vals, idx, scores, k)
            vi = idx[0].cpu().tolist()
            vv = vals[0].cpu().tolist()
            return [(self._ids[i], float(v)) for i, v in zip(vi, vv)][:top_k]
        scores = self._vecs @ query
        idx = np.argsort(-scores)[:top_k]
        return [(self._ids[i], float(scores[i])) for i in idx]

    def state(self):
        return {"ids": self._ids, "vecs": self._vecs}

    def load_state(self, ids, vecs):
        self._ids = list(ids)
        self._vecs = np.asarray(vecs, dtype=np.float32).reshape(len(ids), self.dim)
        self._pos = {d: i for i, d in enumerate(self._ids)}
        self._gpu_vecs = None


class VectorStoreIndex:
    """One named index: docs + vector index (flat local or qdrant remote)
    + BM25 stats."""

    def __init__(self, name: str, embedding, use_gpu: Optional[bool] = None,
                 index_factory=None):
        self.name = name
        self.embedding = embedding
        self.docs: Dict[str, Document] = {}
        self.flat = index_factory(embedding.dim, name) if index_factory \
            else FlatIndex(embedding.dim, use_gpu)
        self.bm25 = BM25Index()
        self.lock = threading.RLock()

    # ---- document ops ----
    def index_documents(self, texts: List[str], metadatas=None) -> List[str]:
        metadatas = metadatas or [{}] * len(texts)
        vecs = self.embedding.embed(texts)
        out = []
        with self.lock:
            for text, meta, vec in zip(texts, metadatas, vecs):
                did = doc_id_for(text)
                self.docs[did] = Document(did, text, meta or {})
                self.flat.add(did, vec)
                self.bm25.add(did, text)
                out.append(did)
        return out

    def update_document(self, doc_id: str, text: str, metadata=None) -> str:
        with self.lock:
            self.delete_document(doc_id)
        return self.index_documents([text], [metadata or {}])[0]

    def delete_document(self, doc_id: str) -> bool:
        with self.lock:
            if doc_id not in self.docs:
                return False
            del self.docs[doc_id]
            self.flat.remove(doc_id)
            self.bm25.remove(doc_id)
            return True

    def list_documents(self, limit: int = 100, offset: int = 0):
        with self.lock:
            items = list(self.docs.values())[offset:offset + limit]
            return [{"doc_id": d.doc_id, "text": d.text,
                     "metadata": d.metadata} for d in items]

    # ---- retrieval ----
    def retrieve(self, query: str, top_k: int = 5,
                 vector_weight: float = 0.7, bm25_weight: float = 0.3
                 ) -> List[RetrievedDoc]:
        with self.lock:
            if not self.docs:
                return []
            qv = self.embedding.embed_query(query)
            vec_hits = self.flat.search(qv, top_k * 3)
            bm_hits = self.bm25.search(query, top_k * 3)
            # weighted fusion: vector scores normalized to [0,1]; bm25
            # contributes reciprocal-rank (reference semantics)
            fused: Dict[str, float] = {}
            if vec_hits:
                smax = max(s for _, s in vec_hits)
                smin = min(s for _, s in vec_hits)
                rng = (smax - smin) or 1.0
                for did, s in vec_hits:
                    fused[did] = fused.get(did, 0.0) + \
                        vector_weight * (s - smin) / rng
            for rank, (did, _) in enumerate(bm_hits):
                fused[did] = fused.get(did, 0.0) + bm25_weight / (1 + rank)
            best = sorted(fused.items(), key=lambda kv: -kv[1])[:top_k]
            return [RetrievedDoc(did, self.docs[did].text, score,
                                 self.docs[did].metadata)
                    for did, score in best if did in self.docs]

    # ---- persistence ----
    def persist(self, path: str) -> None:
        with self.lock:
            os.makedirs(path, exist_ok=True)
            st = self.flat.state()
            np.save(os.path.join(path, "vectors.npy"), st["vecs"])
            with open(os.path.join(path, "docs.json"), "w") as f:
                json.dump({
                    "name": self.name,
                    "ids": st["ids"],
                    "docs": [{"doc_id": d.doc_id, "text": d.text,
                              "metadata": d.metadata}
                             for d in self.docs.values()],
                }, f)

    def load(self, path: str) -> None:
        with self.lock:
            vecs = np.load(os.path.join(path, "vectors.npy"))
            with open(os.path.join(path, "docs.json")) as f:
                meta = json.load(f)
            self.docs = {d["doc_id"]: Document(d["doc_id"], d["text"],
                                               d.get("metadata", {}))
                         for d in meta["docs"]}
            self.flat.load_state(meta["ids"], vecs)
            self.bm25 = BM25Index()
            for d in self.docs.values():
                self.bm25.add(d.doc_id, d.text)


class VectorStoreManager:
    """Named-index manager (reference: VectorStoreManager)."""

    def __init__(self, embedding, use_gpu: Optional[bool] = None,
                 index_factory=None):
        self.embedding = embedding
        self.use_gpu = use_gpu
        self.index_factory = index_factory
        self.indexes: Dict[str, VectorStoreIndex] = {}
        self.lock = threading.RLock()

    def get(self, name: str, create: bool = False) -> VectorStoreIndex:
        with self.lock:
            if name not in self.indexes:
                if not create:
                    raise KeyError(f"index {name!r} not found")
                self.indexes[name] = VectorStoreIndex(
                    name, self.embedding, self.use_gpu, self.index_factory)
            return self.indexes[name]

    def list_indexes(self) -> List[str]:
        with self.lock:
            return sorted(self.indexes)

    def delete_index(self, name: str) -> bool:
        with self.lock:
            return self.indexes.pop(name, None) is not None

    def persist_all(self, root: str) -> List[str]:
        with self.lock:
            out = []
            for name, idx in self.indexes.items():
                idx.persist(os.path.join(root, name))
                out.append(name)
            return out

    def load_index(self, name: str, root: str) -> VectorStoreIndex:
        idx = self.get(name, create=True)
        idx.load(os.path.join(root, name))
        return idx
