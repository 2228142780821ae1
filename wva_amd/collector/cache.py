"""TTL cache for metric results keyed by (query, params).

Parity: reference internal/collector/source/cache.go + cache_value.go —
TTL cache with background cleanup (cleanup runs inline on access here,
avoiding a dedicated goroutine; behavior is equivalent).
"""
from __future__ import annotations

import threading
import time
from typing import Dict, Optional, Tuple

from .source import MetricResult


def cache_key(query: str, params: Dict[str, str]) -> str:
    parts = [query] + [f"{k}={v}" for k, v in sorted(params.items())]
    return "|".join(parts)


class TTLCache:
    def __init__(self, ttl_seconds: float = 30.0, cleanup_interval_seconds: float = 60.0):
        self.ttl_seconds = ttl_seconds
        self.cleanup_interval_seconds = cleanup_interval_seconds
        self._lock = threading.RLock()
        self._entries: Dict[str, Tuple[float, MetricResult]] = {}
        self._last_cleanup = time.monotonic()

    def get(self, key: str) -> Optional[MetricResult]:
        with self._lock:
            self._maybe_cleanup()
            entry = self._entries.get(key)
            if entry is None:
                return None
            stored_at, result = entry
            if time.monotonic() - stored_at > self.ttl_seconds:
                del self._entries[key]
                return None
            return result

    def put(self, key: str, result: MetricResult) -> None:
        with self._lock:
            self._entries[key] = (time.monotonic(), result)

    def invalidate(self, key: Optional[str] = None) -> None:
        with self._lock:
            if key is None:
                self._entries.clear()
            else:
                self._entries.pop(key, None)

    def _maybe_cleanup(self) -> None:
        now = time.monotonic()
        if now - self._last_cleanup < self.cleanup_interval_seconds:
            return
        self._last_cleanup = now
        expired = [
            k
            for k, (stored_at, _) in self._entries.items()
            if now - stored_at > self.ttl_seconds
        ]
        for k in expired:
            del self._entries[k]

    def __len__(self) -> int:
        with self._lock:
            return len(self._entries)
