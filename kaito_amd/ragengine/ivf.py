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
        # GPU-resident search state (vectors + centroids + list-sorted
        # row order), rebuilt lazily after mutations — searches do NO
        # per-query host↔device vector transfers and no Python gather
        self._gpu = None

    # ---- FlatIndex surface -------------------------------------------------
    def __len__(self):
        return len(self._flat)

    @property
    def use_gpu(self):
        return self._flat.use_gpu

    def add(self, doc_id: str, vec: np.ndarray) -> None:
        self._flat.add(doc_id, vec)
        self._gpu = None
        if self.centroids is not None:
            self._assign[doc_id] = int(np.argmax(self.centroids @ vec))
        elif len(self._flat) >= self.min_train:
            self.train()

    def remove(self, doc_id: str) -> None:
        self._flat.remove(doc_id)
        self._assign.pop(doc_id, None)
        self._gpu = None

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
        self._gpu = None
        st = self._flat.state()
        if not st["ids"]:
            self._assign = {}
            return
        a = np.argmax(st["vecs"] @ self.centroids.T, axis=1)
        self._assign = {d: int(j) for d, j in zip(st["ids"], a)}

    # ---- GPU-resident search state ----------------------------------------
    def _gpu_state(self):
        """(vecs [N,D], centroids [nlist,D], order [N] int64 rows sorted
        by inverted list, offsets [nlist+1] host ints) — built once per
        mutation epoch, reused by every search."""
        if self._gpu is not None:
            return self._gpu
        import torch
        st = self._flat.state()
        dev = "cuda"
        vecs = torch.from_numpy(np.ascontiguousarray(st["vecs"])).to(dev)
        cent = torch.from_numpy(self.centroids).to(dev)
        assign = np.array([self._assign.get(d, 0) for d in st["ids"]],
                          dtype=np.int64)
        order_np = np.argsort(assign, kind="stable")
        counts = np.bincount(assign, minlength=self.nlist)
        offsets = np.concatenate([[0], np.cumsum(counts)]).tolist()
        order = torch.from_numpy(order_np).to(dev)
        self._gpu = (vecs, cent, order, offsets)
        return self._gpu

    # ---- search ------------------------------------------------------------
    def search(self, query: np.ndarray, top_k: int):
        if self.centroids is None:
            return self._flat.search(query, top_k)          # flat fallback
        if self._flat.use_gpu:
            return self._search_gpu(query, top_k)
        # CPU fallback
        coarse = self.centroids @ query
        lists = set(np.argpartition(-coarse, self.nprobe - 1)
                    [:self.nprobe].tolist())
        st = self._flat.state()
        cand = [i for i, d in enumerate(st["ids"])
                if self._assign.get(d, -1) in lists]
        if not cand:
            return self._flat.search(query, top_k)
        scores = st["vecs"][cand] @ query
        order = np.argsort(-scores)[:top_k]
        return [(st["ids"][cand[i]], float(scores[i])) for i in order]

    def _search_gpu(self, query: np.ndarray, top_k: int):
        """Fully GPU-resident IVF search: coarse centroid GEMM → HIP
        top-k over lists → device gather of the probed lists' rows →
        fine GEMM → HIP top-k. The only host↔device traffic per query is
        the query vector up and (nprobe + top_k) scalars down."""
        import torch
        from .. import ops
        ops.load_extension()
        vecs, cent, order, offsets = self._gpu_state()
        dev = vecs.device
        q = torch.from_numpy(
            np.ascontiguousarray(query[None, :].astype(np.float32))).to(dev)
        # coarse: [1, nlist] scores → nprobe list ids (HIP top-k)
        coarse = (q @ cent.T).contiguous()
        npb = min(self.nprobe, self.nlist)
        cvals = torch.empty(1, npb, dtype=torch.float32, device=dev)
        cidx = torch.empty(1, npb, dtype=torch.int32, device=dev)
        torch.ops.kaito.topk(cvals, cidx, coarse, npb)
        lists = cidx[0].cpu().tolist()      # nprobe ints — the tiny D2H
        segs = [order[offsets[j]:offsets[j + 1]] for j in lists
                if offsets[j + 1] > offsets[j]]
        if not segs:
            return self._flat.search(query, top_k)
        cand = torch.cat(segs)              # device gather indices
        sub = vecs.index_select(0, cand)    # device row gather
        scores = (q @ sub.T).contiguous()   # MFMA GEMM
        k = min(top_k, cand.numel(), 32)
        vals = torch.empty(1, k, dtype=torch.float32, device=dev)
        idx = torch.empty(1, k, dtype=torch.int32, device=dev)
        torch.ops.kaito.topk(vals, idx, scores, k)
        rows = cand[idx[0].long()].cpu().tolist()
        ids = self._flat.state()["ids"]
        return [(ids[r], float(v))
                for r, v in zip(rows, vals[0].cpu().tolist())][:top_k]
