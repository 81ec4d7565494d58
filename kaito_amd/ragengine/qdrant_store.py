"""Qdrant vector-DB backend — parity with the reference's
presets/ragengine/vector_store/qdrant_store.py (568 L; selected by
VECTOR_DB_TYPE=qdrant + VECTOR_DB_URL). REST client (collections /
points upsert / points search) so no qdrant-client dependency is needed;
plugs in as the FlatIndex replacement inside VectorStoreIndex.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np


class QdrantIndex:
    """Same surface as vector_store.FlatIndex, backed by a Qdrant server."""

    def __init__(self, dim: int, collection: str, url: str,
                 api_key: str = "", timeout: float = 30.0):
        import httpx
        self.dim = dim
        self.collection = collection
        self._client = httpx.Client(
            base_url=url.rstrip("/"), timeout=timeout,
            headers={"api-key": api_key} if api_key else {})
        self._ids: List[str] = []          # insertion order (len support)
        self._num: Dict[str, int] = {}     # doc_id → numeric point id
        self._next = 0
        self._ensure_collection()

    def _ensure_collection(self):
        r = self._client.get(f"/collections/{self.collection}")
        if r.status_code == 200:
            return
        r = self._client.put(f"/collections/{self.collection}", json={
            "vectors": {"size": self.dim, "distance": "Cosine"}})
        r.raise_for_status()

    def __len__(self):
        return len(self._ids)

    def add(self, doc_id: str, vec: np.ndarray) -> None:
        if doc_id not in self._num:
            self._num[doc_id] = self._next
            self._next += 1
            self._ids.append(doc_id)
        pid = self._num[doc_id]
        r = self._client.put(
            f"/collections/{self.collection}/points?wait=true", json={
                "points": [{"id": pid, "vector": vec.tolist(),
                            "payload": {"doc_id": doc_id}}]})
        r.raise_for_status()

    def remove(self, doc_id: str) -> None:
        pid = self._num.pop(doc_id, None)
        if pid is None:
            return
        self._ids.remove(doc_id)
        self._client.post(
            f"/collections/{self.collection}/points/delete?wait=true",
            json={"points": [pid]})

    def search(self, query: np.ndarray, top_k: int
               ) -> List[Tuple[str, float]]:
        if not self._ids:
            return []
        r = self._client.post(
            f"/collections/{self.collection}/points/search", json={
                "vector": query.tolist(), "limit": int(top_k),
                "with_payload": True})
        r.raise_for_status()
        out = []
        for hit in r.json().get("result", []):
            did = (hit.get("payload") or {}).get("doc_id")
            if did is not None:
                out.append((did, float(hit.get("score", 0.0))))
        return out

    # persistence is server-side for Qdrant; keep interface compatibility
    def state(self):
        return {"ids": list(self._ids), "vecs": np.zeros((0, self.dim))}

    def load_state(self, ids, vecs):
        for i, d in enumerate(ids):
            if d not in self._num:
                self._ids.append(d)
                self._num[d] = self._next
                self._next += 1


def make_index(cfg, dim: int, name: str):
    """Backend factory (reference: VECTOR_DB_TYPE faiss|qdrant)."""
    if cfg.vector_db_type.lower() == "qdrant" and cfg.vector_db_url:
        return QdrantIndex(dim, name, cfg.vector_db_url,
                           cfg.vector_db_access_secret)
    from .vector_store import FlatIndex
    return FlatIndex(dim)
