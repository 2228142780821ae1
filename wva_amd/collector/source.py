"""Metrics source abstraction.

Parity: reference internal/collector/source/source.go:14-130 —
MetricsSource{QueryList, Refresh(RefreshSpec), Get}, MetricValue with
timestamp + labels and staleness helpers, MetricResult with per-query error.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Protocol


@dataclass
class MetricValue:
    value: float = 0.0
    timestamp: float = 0.0  # unix seconds
    labels: Dict[str, str] = field(default_factory=dict)

    def age_seconds(self, now: Optional[float] = None) -> float:
        if self.timestamp <= 0:
            return 0.0
        return max((now if now is not None else time.time()) - self.timestamp, 0.0)

    def is_stale(self, threshold_seconds: float) -> bool:
        return self.age_seconds() > threshold_seconds


@dataclass
class MetricResult:
    query: str = ""
    values: List[MetricValue] = field(default_factory=list)
    error: Optional[Exception] = None
    fetched_at: float = 0.0

    def has_error(self) -> bool:
        return self.error is not None

    def first_value(self) -> Optional[MetricValue]:
        return self.values[0] if self.values else None


@dataclass
class RefreshSpec:
    queries: List[str] = field(default_factory=list)
    params: Dict[str, str] = field(default_factory=dict)


class MetricsSource(Protocol):
    def name(self) -> str: ...

    def query_list(self) -> "QueryList": ...  # noqa: F821

    def refresh(self, spec: RefreshSpec) -> Dict[str, MetricResult]: ...

    def get(self, query: str, params: Dict[str, str]) -> Optional[MetricResult]: ...
