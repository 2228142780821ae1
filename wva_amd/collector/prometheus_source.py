"""Prometheus-backed metrics source.

Parity: reference internal/collector/source/prometheus/prometheus_source.go
:24-302 — concurrent query fan-out (thread per query), per-query 10s
timeout, retry with backoff, vector/scalar/matrix parsing with NaN→0,
30s TTL cache.

Talks to the Prometheus HTTP API (`/api/v1/query`) via `requests`.
"""
from __future__ import annotations

import concurrent.futures
import math
import threading
import time
from typing import Dict, List, Optional

import requests

from ..utils.backoff import retry_with_backoff
from ..utils.logging import get_logger
from .cache import TTLCache, cache_key
from .query_template import QUERY_TYPE_PROMQL, QueryList
from .registry import PROMETHEUS_SOURCE_NAME
from .source import MetricResult, MetricValue, RefreshSpec

log = get_logger("collector.prometheus")

DEFAULT_QUERY_TIMEOUT_SECONDS = 10.0
DEFAULT_CACHE_TTL_SECONDS = 30.0


def _nan_to_zero(v: float) -> float:
    if math.isnan(v) or math.isinf(v):
        return 0.0
    return v


def parse_prometheus_response(data: dict) -> List[MetricValue]:
    """Parse a /api/v1/query JSON body into MetricValues.

    Handles vector, scalar and matrix result types; NaN/Inf sample values
    become 0 (reference prometheus_source.go:171-250).
    """
    if data.get("status") != "success":
        raise RuntimeError(f"prometheus error: {data.get('error', 'unknown')}")
    result = data.get("data", {})
    rtype = result.get("resultType")
    raw = result.get("result", [])
    values: List[MetricValue] = []
    if rtype == "vector":
        for sample in raw:
            ts, val = sample.get("value", [0, "0"])
            labels = dict(sample.get("metric", {}))
            values.append(
                MetricValue(
                    value=_nan_to_zero(float(val)),
                    timestamp=float(ts),
                    labels=labels,
                )
            )
    elif rtype == "scalar":
        ts, val = raw
        values.append(
            MetricValue(value=_nan_to_zero(float(val)), timestamp=float(ts))
        )
    elif rtype == "matrix":
        for series in raw:
            labels = dict(series.get("metric", {}))
            series_values = series.get("values", [])
            if series_values:
                ts, val = series_values[-1]  # most recent sample
                values.append(
                    MetricValue(
                        value=_nan_to_zero(float(val)),
                        timestamp=float(ts),
                        labels=labels,
                    )
                )
    elif rtype is not None:
        raise RuntimeError(f"unsupported result type: {rtype}")
    return values


class PrometheusSource:
    def __init__(
        self,
        base_url: str,
        query_timeout_seconds: float = DEFAULT_QUERY_TIMEOUT_SECONDS,
        cache_ttl_seconds: float = DEFAULT_CACHE_TTL_SECONDS,
        bearer_token: str = "",
        token_path: str = "",
        verify_tls: bool = True,
        ca_cert_path: str = "",
        client_cert_path: str = "",
        client_key_path: str = "",
        session: Optional[requests.Session] = None,
        max_retries: int = 3,
    ):
        """TLS parity with reference internal/utils/tls.go: CA bundle,
        mTLS client cert/key, bearer token (inline or mounted file)."""
        self.base_url = base_url.rstrip("/")
        self.query_timeout_seconds = query_timeout_seconds
        self._query_list = QueryList()
        self._cache = TTLCache(ttl_seconds=cache_ttl_seconds)
        self._session = session or requests.Session()
        if not bearer_token and token_path:
            try:
                with open(token_path) as f:
                    bearer_token = f.read().strip()
            except OSError:
                pass
        self._bearer_token = bearer_token
        # requests `verify` accepts False or a CA bundle path
        self._verify_tls = (
            ca_cert_path if (verify_tls and ca_cert_path) else verify_tls
        )
        if client_cert_path and client_key_path:
            self._session.cert = (client_cert_path, client_key_path)
        self._max_retries = max_retries
        self._lock = threading.Lock()

    def name(self) -> str:
        return PROMETHEUS_SOURCE_NAME

    def query_list(self) -> QueryList:
        return self._query_list

    # --- HTTP ---

    def _execute_query(self, promql: str) -> List[MetricValue]:
        headers = {}
        if self._bearer_token:
            headers["Authorization"] = f"Bearer {self._bearer_token}"

        def attempt():
            resp = self._session.get(
                f"{self.base_url}/api/v1/query",
                params={"query": promql},
                headers=headers,
                timeout=self.query_timeout_seconds,
                verify=self._verify_tls,
            )
            resp.raise_for_status()
            return parse_prometheus_response(resp.json())

        return retry_with_backoff(
            attempt, max_attempts=self._max_retries, initial_delay=0.1, max_delay=2.0
        )

    def validate(self) -> bool:
        """Connectivity check (reference utils.ValidatePrometheusAPI)."""
        try:
            resp = self._session.get(
                f"{self.base_url}/api/v1/query",
                params={"query": "vector(1)"},
                timeout=self.query_timeout_seconds,
                verify=self._verify_tls,
            )
            resp.raise_for_status()
            return resp.json().get("status") == "success"
        except Exception:  # noqa: BLE001
            return False

    # --- MetricsSource ---

    def refresh(self, spec: RefreshSpec) -> Dict[str, MetricResult]:
        """Fan out all queries concurrently; per-query errors are carried
        in MetricResult.error (never raised) so one failed query family
        doesn't kill the tick."""
        results: Dict[str, MetricResult] = {}

        def run_one(name: str) -> MetricResult:
            key = cache_key(name, spec.params)
            cached = self._cache.get(key)
            if cached is not None:
                return cached
            try:
                template = self._query_list.get(name)
                if template.type != QUERY_TYPE_PROMQL:
                    raise ValueError(
                        f"prometheus source only supports promql queries, "
                        f"got {template.type} for {name}"
                    )
                promql = template.render(spec.params)
                values = self._execute_query(promql)
                result = MetricResult(
                    query=name, values=values, fetched_at=time.time()
                )
                self._cache.put(key, result)
                return result
            except Exception as e:  # noqa: BLE001 — per-query error captured
                return MetricResult(query=name, error=e, fetched_at=time.time())

        if len(spec.queries) == 1:
            results[spec.queries[0]] = run_one(spec.queries[0])
            return results

        with concurrent.futures.ThreadPoolExecutor(
            max_workers=max(len(spec.queries), 1)
        ) as pool:
            futures = {pool.submit(run_one, q): q for q in spec.queries}
            for fut in concurrent.futures.as_completed(futures):
                results[futures[fut]] = fut.result()
        return results

    def get(self, query: str, params: Dict[str, str]) -> Optional[MetricResult]:
        return self._cache.get(cache_key(query, params))


def format_prometheus_duration(seconds: float) -> str:
    """Seconds → Prometheus duration string ("600s" style, exact)."""
    if seconds == int(seconds):
        s = int(seconds)
        if s % 3600 == 0:
            return f"{s // 3600}h"
        if s % 60 == 0:
            return f"{s // 60}m"
        return f"{s}s"
    return f"{seconds:.3f}s"
