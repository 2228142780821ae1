"""Pod-scraping metrics source (EPP endpoint-picker pods).

Parity: reference internal/collector/source/pod/pod_scraping_source.go
:29-388 — discovers ready pods behind the EPP Service selector, scrapes
each pod's metrics endpoint concurrently (max 10 in flight, 5s timeout,
optional bearer token), parses Prometheus text exposition and serves all
samples under the single query name `all_metrics` with `pod` and
`__name__` labels attached.
"""
from __future__ import annotations

import concurrent.futures
import re
import time
from typing import Callable, Dict, List, Optional

import requests

from ..kube.fake import FakeCluster
from ..kube.objects import EndpointPool, Pod
from ..utils.logging import get_logger
from .query_template import QUERY_TYPE_METRIC_NAME, QueryList, QueryTemplate
from .source import MetricResult, MetricValue, RefreshSpec

log = get_logger("collector.pod_scrape")

ALL_METRICS_QUERY = "all_metrics"
DEFAULT_SCRAPE_TIMEOUT_SECONDS = 5.0
DEFAULT_MAX_CONCURRENT_SCRAPES = 10

_SAMPLE_RE = re.compile(
    r'^(?P<name>[a-zA-Z_:][a-zA-Z0-9_:]*)'
    # label block: quoted strings are consumed atomically so a } (or
    # anything else) INSIDE a label value cannot terminate the block
    r'(?:\{(?P<labels>(?:[^"}]|"(?:[^"\\]|\\.)*")*)\})?'
    r'\s+(?P<value>[^\s]+)'
    r'(?:\s+(?P<ts>\d+))?$'
)
_LABEL_RE = re.compile(r'(\w+)="((?:[^"\\]|\\.)*)"')


def _unescape_label_value(raw: str) -> str:
    """Exposition-format label unescape (\\\\, \\", \\n) in ONE
    left-to-right pass — sequential str.replace corrupts sequences like
    ``\\\\n`` (escaped backslash + literal n), turning them into a
    newline."""
    out = []
    i, n = 0, len(raw)
    while i < n:
        ch = raw[i]
        if ch == "\\" and i + 1 < n:
            nxt = raw[i + 1]
            if nxt == "n":
                out.append("\n")
            elif nxt in ('"', "\\"):
                out.append(nxt)
            else:
                out.append(ch)
                out.append(nxt)
            i += 2
        else:
            out.append(ch)
            i += 1
    return "".join(out)


def parse_prometheus_text(text: str) -> List[MetricValue]:
    """Parse Prometheus text exposition format into MetricValues with
    __name__ labels."""
    now = time.time()
    values: List[MetricValue] = []
    # split on \n ONLY: the exposition format is newline-delimited, and
    # str.splitlines() would also break lines at \x0b/\x1c-\x1e/ …,
    # which are legal raw bytes inside a label value
    for line in text.split("\n"):
        line = line.strip()
        if not line or line.startswith("#"):
            continue
        m = _SAMPLE_RE.match(line)
        if not m:
            continue
        labels = {"__name__": m.group("name")}
        raw_labels = m.group("labels")
        if raw_labels:
            for lm in _LABEL_RE.finditer(raw_labels):
                labels[lm.group(1)] = _unescape_label_value(lm.group(2))
        try:
            value = float(m.group("value"))
        except ValueError:
            continue
        ts = m.group("ts")
        timestamp = float(ts) / 1000.0 if ts else now
        values.append(MetricValue(value=value, timestamp=timestamp, labels=labels))
    return values


# (url, headers, timeout) → response text; pluggable for tests/emulation
FetchFunc = Callable[[str, Dict[str, str], float], str]


def _default_fetch(url: str, headers: Dict[str, str], timeout: float) -> str:
    resp = requests.get(url, headers=headers, timeout=timeout)
    resp.raise_for_status()
    return resp.text


class PodScrapingSource:
    """One instance per EndpointPool (InferencePool)."""

    def __init__(
        self,
        cluster: FakeCluster,
        pool: EndpointPool,
        bearer_token: str = "",
        metrics_reader_secret_name: str = "",
        metrics_reader_secret_key: str = "token",
        scrape_timeout_seconds: float = DEFAULT_SCRAPE_TIMEOUT_SECONDS,
        max_concurrent_scrapes: int = DEFAULT_MAX_CONCURRENT_SCRAPES,
        fetch: Optional[FetchFunc] = None,
        metrics_path: str = "/metrics",
        metrics_scheme: str = "http",
    ):
        self.cluster = cluster
        self.pool = pool
        self.bearer_token = bearer_token
        self.metrics_reader_secret_name = metrics_reader_secret_name
        self.metrics_reader_secret_key = metrics_reader_secret_key
        self.scrape_timeout_seconds = scrape_timeout_seconds
        self.max_concurrent_scrapes = max(1, max_concurrent_scrapes)
        self.fetch = fetch or _default_fetch
        self.metrics_path = metrics_path
        self.metrics_scheme = metrics_scheme
        self._query_list = QueryList()
        self._query_list.must_register(QueryTemplate(
            name=ALL_METRICS_QUERY,
            type=QUERY_TYPE_METRIC_NAME,
            template="",
            params=[],
            description="All metrics scraped from EPP pods behind the pool service",
        ))
        self._last_result: Optional[MetricResult] = None

    def name(self) -> str:
        return f"pool/{self.pool.namespace}/{self.pool.name}"

    def query_list(self) -> QueryList:
        return self._query_list

    # --- pod discovery via EPP service selector ---

    def _discover_pods(self) -> List[Pod]:
        picker = self.pool.endpoint_picker
        svc = self.cluster.try_get("Service", picker.namespace, picker.service_name)
        if svc is None:
            return []
        pods = self.cluster.list("Pod", namespace=picker.namespace)
        ready = []
        for pod in pods:
            if not pod.is_ready():
                continue
            if svc.selector and all(
                pod.metadata.labels.get(k) == v for k, v in svc.selector.items()
            ):
                ready.append(pod)
        return ready

    def _auth_token(self) -> str:
        """Token resolution per reference getAuthToken
        (pod_scraping_source.go:300-331): explicit token wins; else read
        the metrics-reader Secret from the EPP service namespace (fresh
        read each refresh — live rotation); a missing Secret or key
        means auth is OPTIONAL, not an error."""
        if self.bearer_token:
            return self.bearer_token
        if not self.metrics_reader_secret_name:
            return ""
        secret = self.cluster.try_get(
            "Secret",
            self.pool.endpoint_picker.namespace,
            self.metrics_reader_secret_name,
        )
        if secret is None:
            return ""
        return secret.data.get(self.metrics_reader_secret_key, "")

    def _scrape_pod(self, pod: Pod, token: str) -> List[MetricValue]:
        picker = self.pool.endpoint_picker
        host = pod.status.pod_ip or pod.name
        url = (
            f"{self.metrics_scheme}://{host}:"
            f"{picker.metrics_port_number}{self.metrics_path}"
        )
        headers = {}
        if token:
            headers["Authorization"] = f"Bearer {token}"
        text = self.fetch(url, headers, self.scrape_timeout_seconds)
        values = parse_prometheus_text(text)
        for v in values:
            v.labels["pod"] = pod.name
        return values

    # --- MetricsSource ---

    def refresh(self, spec: RefreshSpec) -> Dict[str, MetricResult]:
        pods = self._discover_pods()
        token = self._auth_token()
        all_values: List[MetricValue] = []
        errors: List[Exception] = []
        if pods:
            with concurrent.futures.ThreadPoolExecutor(
                max_workers=min(self.max_concurrent_scrapes, len(pods))
            ) as pool:
                futures = {
                    pool.submit(self._scrape_pod, p, token): p for p in pods
                }
                for fut in concurrent.futures.as_completed(futures):
                    try:
                        all_values.extend(fut.result())
                    except Exception as e:  # noqa: BLE001 — per-pod failure tolerated
                        errors.append(e)
                        log.debug(
                            "scrape failed for pod %s: %s", futures[fut].name, e
                        )
        result = MetricResult(
            query=ALL_METRICS_QUERY,
            values=all_values,
            fetched_at=time.time(),
            # All pods failing (with pods present) is an error condition
            error=errors[0] if errors and not all_values else None,
        )
        self._last_result = result
        return {ALL_METRICS_QUERY: result}

    def get(self, query: str, params: Dict[str, str]) -> Optional[MetricResult]:
        if query != ALL_METRICS_QUERY:
            return None
        return self._last_result
