"""Source registry: name → MetricsSource.

Parity: reference internal/collector/source/registry.go — "prometheus"
plus one PodScrapingSource per InferencePool.
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional

from .source import MetricsSource

PROMETHEUS_SOURCE_NAME = "prometheus"


class SourceRegistry:
    def __init__(self) -> None:
        self._lock = threading.RLock()
        self._sources: Dict[str, MetricsSource] = {}

    def register(self, source: MetricsSource) -> None:
        with self._lock:
            self._sources[source.name()] = source

    def get(self, name: str) -> Optional[MetricsSource]:
        with self._lock:
            return self._sources.get(name)

    def unregister(self, name: str) -> None:
        with self._lock:
            self._sources.pop(name, None)

    def names(self) -> List[str]:
        with self._lock:
            return sorted(self._sources)
