"""Controller expectations — the informer-cache staleness guard
(reference: pkg/utils/controller.go ControllerExpectations, used by the
InferenceSet controller at inferenceset_controller.go:336-386 and the
Workspace controller at :351).

A reconcile that creates or deletes children records how many it expects
to see materialize; until the watch cache catches up (`satisfied()`),
subsequent reconciles must not act on child counts — otherwise a stale
list triggers duplicate creates or over-deletes. Creations/deletions
observed from the store tick the counters back down; expectations also
expire (default 5 min) so a lost watch event cannot wedge a controller.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from threading import Lock
from typing import Dict, Optional

EXPECTATION_TIMEOUT_S = 300.0


@dataclass
class _Exp:
    adds: int = 0
    dels: int = 0
    stamp: float = field(default_factory=time.monotonic)

    def fulfilled(self) -> bool:
        return self.adds <= 0 and self.dels <= 0

    def expired(self) -> bool:
        return time.monotonic() - self.stamp > EXPECTATION_TIMEOUT_S


class ControllerExpectations:
    def __init__(self):
        self._by_key: Dict[str, _Exp] = {}
        self._lock = Lock()

    def expect_creations(self, key: str, n: int) -> None:
        with self._lock:
            e = self._by_key.setdefault(key, _Exp())
            e.adds += n
            e.stamp = time.monotonic()

    def expect_deletions(self, key: str, n: int) -> None:
        with self._lock:
            e = self._by_key.setdefault(key, _Exp())
            e.dels += n
            e.stamp = time.monotonic()

    def creation_observed(self, key: str) -> None:
        with self._lock:
            e = self._by_key.get(key)
            if e is not None:
                e.adds -= 1

    def deletion_observed(self, key: str) -> None:
        with self._lock:
            e = self._by_key.get(key)
            if e is not None:
                e.dels -= 1

    def satisfied(self, key: str) -> bool:
        """True when it is safe to act on the listed child set."""
        with self._lock:
            e = self._by_key.get(key)
            if e is None:
                return True
            if e.fulfilled() or e.expired():
                del self._by_key[key]
                return True
            return False

    def delete(self, key: str) -> None:
        with self._lock:
            self._by_key.pop(key, None)

    def pending(self, key: str) -> Optional[tuple]:
        with self._lock:
            e = self._by_key.get(key)
            return None if e is None else (e.adds, e.dels)
