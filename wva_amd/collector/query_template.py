"""Query template registry with {{.param}} substitution + PromQL escaping.

Parity: reference internal/collector/source/query_template.go — named
query registry, QueryTypePromQL vs QueryTypeMetricName, parameter
validation, and label-value escaping to prevent PromQL injection.
"""
from __future__ import annotations

import re
import threading
from dataclasses import dataclass, field
from typing import Dict, List

# Common parameter names
PARAM_NAMESPACE = "namespace"
PARAM_MODEL_ID = "modelID"
PARAM_POD_FILTER = "podFilter"
PARAM_RETENTION_PERIOD = "retentionPeriod"

QUERY_TYPE_METRIC_NAME = "metric"
QUERY_TYPE_PROMQL = "promql"

_PLACEHOLDER_RE = re.compile(r"\{\{\.(\w+)\}\}")


def escape_promql_value(value: str) -> str:
    """Escape backslashes then double quotes for PromQL label matchers."""
    return value.replace("\\", "\\\\").replace('"', '\\"')


@dataclass
class QueryTemplate:
    name: str
    type: str = QUERY_TYPE_PROMQL
    template: str = ""
    params: List[str] = field(default_factory=list)
    description: str = ""

    def render(self, params: Dict[str, str]) -> str:
        missing = [p for p in self.params if p not in params]
        if missing:
            raise KeyError(
                f"query {self.name!r} missing required params: {missing}"
            )

        def sub(m: re.Match) -> str:
            key = m.group(1)
            if key not in params:
                raise KeyError(f"query {self.name!r}: unknown placeholder {key!r}")
            return escape_promql_value(params[key])

        return _PLACEHOLDER_RE.sub(sub, self.template)


class QueryList:
    def __init__(self) -> None:
        self._lock = threading.RLock()
        self._queries: Dict[str, QueryTemplate] = {}

    def register(self, template: QueryTemplate) -> None:
        with self._lock:
            if template.name in self._queries:
                raise ValueError(f"query {template.name!r} already registered")
            self._queries[template.name] = template

    def must_register(self, template: QueryTemplate) -> None:
        self.register(template)

    def get(self, name: str) -> QueryTemplate:
        with self._lock:
            t = self._queries.get(name)
            if t is None:
                raise KeyError(f"query {name!r} not registered")
            return t

    def has(self, name: str) -> bool:
        with self._lock:
            return name in self._queries

    def names(self) -> List[str]:
        with self._lock:
            return sorted(self._queries)
