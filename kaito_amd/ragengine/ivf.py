"""IVF-flat vector index — the north-star replacement for FAISS IVF
(reference default backend is IndexFlatL2 inside an IndexIDMap,
presets/ragengine/vector_store/faiss_store.py:25-50; IVF is the
coarse-quantized upgrade for >100k-doc corpora).

MI355X mapping: every distance computation is a GEMM (q @ X^T →
hipBLASLt on MFMA) and every selection is the in-tree HIP top-k kernel
(ops/csrc/topk.hip) — same compute path as FlatIndex, restricted to the
nprobe nearest inverted lists. k-means training runs the assignment step
as one [N, nlist] GEMM per iteration on the same path. CPU fallback is
numpy throughout (tests).
"""
from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from .vector_store import FlatIndex


class IVFFlatIndex:
    """Same surface as vector_store.FlatIndex. Untrained (< min_train
    vectors) it scans flat; once trained, search visits the nprobe nearest
    lists only."""

    def __init__(self, dim: int, nlist: int = 64, nprobe: int = 8,
                 min_train: int = 256, use_gpu: Optional[bool] = None,
                 seed: int = 0):
        self.dim = dim
        self.nlist = nlist
        self.nprobe = max(1, min(nprobe, nlist))
        self.min_train = max(min_train, nlist)
        self.seed = seed
        self._flat = FlatIndex(dim, use_gpu)   # master storage + GPU scoring
        self.centroids: Optional[np.ndarray] = None     # [nlist, dim]
        self._assign: Dict[str, int] = {}               # doc_id → list id

    # ---- FlatIndex surface -------------------------------------------------
    def __len__(self):
        return len(self._flat)

    @property
    def use_gpu(self):
        return self._flat.use_gpu

    def add(self, doc_id: str, vec: np.ndarray) -> None:
        self._flat.add(doc_id, vec)
        if self.centroids is not None:
            self._assign[doc_id] = int(np.argmax(self.centroids @ vec))
        elif len(self._flat) >= self.min_train:
            self.train()

    def remove(self, doc_id: str) -> None:
        self._flat.remove(doc_id)
        self._assign.pop(doc_id, None)

    def state(self):
        st = self._flat.state()
        st["centroids"] = self.centroids
        return st

    def load_state(self, ids, vecs, centroids=None):
        self._flat.load_state(ids, vecs)
        self.centroids = None if centroids is None \
            else np.asarray(centroids, dtype=np.float32)
        self._assign = {}
        if self.centroids is None and len(self._flat) >= self.min_train:
            self.train()
        elif self.centroids is not None:
            self._reassign_all()

    # ---- training ----------------------------------------------------------
    def train(self, iters: int = 10) -> None:
        """Spherical k-means over the stored vectors. Assignment is one
        [N, nlist] inner-product GEMM per iteration (vectors normalized →
        cosine); update re-normalizes the mean."""
        X = self._flat.state()["vecs"]
        n = X.shape[0]
        if n < self.nlist:
            return
        rng = np.random.default_rng(self.seed)
        C = X[rng.choice(n, self.nlist, replace=False)].copy()
        for _ in range(iters):
            a = np.argmax(X @ C.T, axis=1)                  # GEMM assign
            for j in range(self.nlist):
                m = a == j
                if m.any():
                    c = X[m].mean(axis=0)
                else:                                       # dead centroid
                    c = X[rng.integers(n)]
                norm = np.linalg.norm(c) or 1.0
                C[j] = c / norm
        self.centroids = C.astype(np.float32)
        self._reassign_all()

    def _reassign_all(self):
        st = self._flat.state()
        if not st["ids"]:
            self._assign = {}
            return
        a = np.argmax(st["vecs"] @ self.centroids.T, axis=1)
        self._assign = {d: int(j) for d, j in zip(st["ids"], a)}

    # ---- search ------------------------------------------------------------
    def search(self, query: np.ndarray, top_k: int):
        if self.centroids is None:
            return self._flat.search(query, top_k)          # flat fallback
        # coarse quantizer: [nlist] scores → nprobe lists
        coarse = self.centroids @ query
        lists = set(np.argpartition(-coarse, self.nprobe - 1)
                    [:self.nprobe].tolist())
        st = self._flat.state()
        cand = [i for i, d in enumerate(st["ids"])
                if self._assign.get(d, -1) in lists]
        if not cand:
            return self._flat.search(query, top_k)
        if self._flat.use_gpu:
            import torch
            from .. import ops
            dev = "cuda"
            sub = torch.from_numpy(st["vecs"][cand]).to(dev)
            q = torch.from_numpy(
                np.ascontiguousarray(query[None, :])).to(dev)
            scores = (q @ sub.T).float().contiguous()       # MFMA GEMM
            k = min(top_k, len(cand), 32)
            vals = torch.empty(1, k, dtype=torch.float32, device=dev)
            idx = torch.empty(1, k, dtype=torch.int32, device=dev)
            ops.load_extension()
            torch.ops.kaito.topk(vals, idx, scores, k)      # HIP top-k
            pairs = [(st["ids"][cand[i]], float(v))
                     for i, v in zip(idx[0].cpu().tolist(),
                                     vals[0].cpu().tolist())]
            return pairs[:top_k]
        scores = st["vecs"][cand] @ query
        order = np.argsort(-scores)[:top_k]
        return [(st["ids"][cand[i]], float(scores[i])) for i in order]
