"""Engine ↔ controller bridge: DecisionCache + DecisionTrigger.

Parity: reference internal/engines/common/cache.go:14-56 — the engine
computes decisions and pre-emits metrics; the reconciler is the only writer
of VA status through the API, fed via this cache + a buffered trigger
channel (cap 1000). Splitting the two keeps API writes out of the engine
loop and survives leader transitions.
"""
from __future__ import annotations

import queue
import threading
from typing import Dict, Optional

from ..analyzers.interfaces import VariantDecision
from ..api.types import OptimizedAlloc

DECISION_TRIGGER_CAPACITY = 1000


class DecisionCache:
    """RWMutex map keyed `namespace/name` → VariantDecision."""

    def __init__(self) -> None:
        self._lock = threading.RLock()
        self._decisions: Dict[str, VariantDecision] = {}

    @staticmethod
    def key(namespace: str, name: str) -> str:
        return f"{namespace}/{name}"

    def set(self, namespace: str, name: str, decision: VariantDecision) -> None:
        with self._lock:
            self._decisions[self.key(namespace, name)] = decision

    def get(self, namespace: str, name: str) -> Optional[VariantDecision]:
        with self._lock:
            return self._decisions.get(self.key(namespace, name))

    def delete(self, namespace: str, name: str) -> None:
        with self._lock:
            self._decisions.pop(self.key(namespace, name), None)

    def __len__(self) -> int:
        with self._lock:
            return len(self._decisions)


class DecisionTrigger:
    """Buffered channel of generic events (va namespace/name keys) feeding
    the reconciler work queue. Drops events when full (same as a full Go
    channel with a non-blocking send)."""

    def __init__(self, capacity: int = DECISION_TRIGGER_CAPACITY):
        self._q: "queue.Queue[str]" = queue.Queue(maxsize=capacity)

    def push(self, namespace: str, name: str) -> bool:
        try:
            self._q.put_nowait(f"{namespace}/{name}")
            return True
        except queue.Full:
            return False

    def pop(self, timeout: Optional[float] = None) -> Optional[str]:
        try:
            return self._q.get(timeout=timeout) if timeout else self._q.get_nowait()
        except queue.Empty:
            return None

    def __len__(self) -> int:
        return self._q.qsize()


def decision_to_optimized_alloc(decision: VariantDecision) -> OptimizedAlloc:
    return OptimizedAlloc(
        last_run_time=decision.last_run_time,
        accelerator=decision.accelerator_name,
        num_replicas=decision.target_replicas,
    )
