"""Fixed-window rolling average with last-updated tracking for eviction.

Parity: reference internal/engines/analyzers/saturation_v2/history.go:8-45.
"""
from __future__ import annotations

import time
from collections import deque
from typing import Deque


class RollingAverage:
    def __init__(self, window_size: int) -> None:
        if window_size <= 0:
            raise ValueError("window_size must be positive")
        self._values: Deque[float] = deque(maxlen=window_size)
        self.last_updated: float = 0.0

    def add(self, value: float) -> None:
        self._values.append(value)
        self.last_updated = time.monotonic()

    def average(self) -> float:
        if not self._values:
            return 0.0
        return sum(self._values) / len(self._values)

    def __len__(self) -> int:
        return len(self._values)
