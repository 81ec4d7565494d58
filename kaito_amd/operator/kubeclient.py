"""Kubernetes client abstraction + in-memory fake.

The reference tests its controllers against a hand-rolled MockClient
(pkg/utils/test/mock_client.go:33-244) rather than envtest; we keep that
strategy: reconcilers depend only on this interface, FakeKubeClient backs
unit tests, and a real impl can wrap the `kubernetes` package when present.
Objects are plain dicts with apiVersion/kind/metadata/spec/status.
"""
from __future__ import annotations

import copy
import threading
from typing import Any, Dict, List, Optional, Tuple

Obj = Dict[str, Any]


class NotFound(KeyError):
    pass


class Conflict(RuntimeError):
    pass


def obj_key(obj: Obj) -> Tuple[str, str, str]:
    md = obj.get("metadata", {})
    return (obj.get("kind", ""), md.get("namespace", ""), md.get("name", ""))


class KubeClient:
    def get(self, kind: str, namespace: str, name: str) -> Obj:
        raise NotImplementedError

    def list(self, kind: str, namespace: Optional[str] = None,
             label_selector: Optional[Dict[str, str]] = None) -> List[Obj]:
        raise NotImplementedError

    def create(self, obj: Obj) -> Obj:
        raise NotImplementedError

    def update(self, obj: Obj) -> Obj:
        raise NotImplementedError

    def delete(self, kind: str, namespace: str, name: str) -> None:
        raise NotImplementedError

    def update_status(self, obj: Obj) -> Obj:
        raise NotImplementedError

    def apply(self, obj: Obj) -> Obj:
        """create-or-update convenience."""
        k, ns, nm = obj_key(obj)
        try:
            existing = self.get(k, ns, nm)
        except NotFound:
            return self.create(obj)
        merged = copy.deepcopy(existing)
        merged["spec"] = obj.get("spec", merged.get("spec"))
        md = merged.setdefault("metadata", {})
        for field in ("labels", "annotations"):
            if field in obj.get("metadata", {}):
                md[field] = obj["metadata"][field]
        return self.update(merged)


class FakeKubeClient(KubeClient):
    def __init__(self):
        self._store: Dict[Tuple[str, str, str], Obj] = {}
        self._rv = 0
        self._lock = threading.RLock()
        self.actions: List[Tuple[str, Tuple[str, str, str]]] = []

    def _bump(self, obj: Obj) -> Obj:
        self._rv += 1
        obj.setdefault("metadata", {})["resourceVersion"] = str(self._rv)
        return obj

    def get(self, kind, namespace, name):
        with self._lock:
            key = (kind, namespace, name)
            if key not in self._store:
                raise NotFound(f"{kind} {namespace}/{name} not found")
            return copy.deepcopy(self._store[key])

    def list(self, kind, namespace=None, label_selector=None):
        with self._lock:
            out = []
            for (k, ns, nm), obj in self._store.items():
                if k != kind:
                    continue
                if namespace is not None and ns != namespace:
                    continue
                if label_selector:
                    labels = obj.get("metadata", {}).get("labels", {})
                    if not all(labels.get(a) == b
                               for a, b in label_selector.items()):
                        continue
                out.append(copy.deepcopy(obj))
            return sorted(out, key=lambda o: o["metadata"]["name"])

    def create(self, obj):
        with self._lock:
            key = obj_key(obj)
            if key in self._store:
                raise Conflict(f"{key} already exists")
            stored = self._bump(copy.deepcopy(obj))
            self._store[key] = stored
            self.actions.append(("create", key))
            return copy.deepcopy(stored)

    def update(self, obj):
        with self._lock:
            key = obj_key(obj)
            if key not in self._store:
                raise NotFound(f"{key} not found")
            stored = self._bump(copy.deepcopy(obj))
            self._store[key] = stored
            self.actions.append(("update", key))
            return copy.deepcopy(stored)

    def update_status(self, obj):
        with self._lock:
            key = obj_key(obj)
            if key not in self._store:
                raise NotFound(f"{key} not found")
            self._store[key]["status"] = copy.deepcopy(obj.get("status", {}))
            self.actions.append(("status", key))
            return copy.deepcopy(self._store[key])

    def delete(self, kind, namespace, name):
        with self._lock:
            key = (kind, namespace, name)
            if key not in self._store:
                raise NotFound(f"{key} not found")
            del self._store[key]
            self.actions.append(("delete", key))
