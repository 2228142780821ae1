"""Collector tests: query templates, TTL cache, Prometheus source against a
local fake Prometheus HTTP API, pod scraping, replica-metrics merge.

Mirrors reference prometheus_source_test.go (450 LoC) /
pod_scraping_source_test.go (956) / replica_metrics coverage.
"""
import json
import threading
import time
from http.server import BaseHTTPRequestHandler, HTTPServer
from urllib.parse import parse_qs, urlparse

import pytest

from wva_amd.api.types import (
    CrossVersionObjectReference,
    ObjectMeta,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.collector.cache import TTLCache, cache_key
from wva_amd.collector.pod_scraping_source import (
    PodScrapingSource,
    parse_prometheus_text,
)
from wva_amd.collector.pod_va_mapper import PodVAMapper
from wva_amd.collector.prometheus_source import (
    PrometheusSource,
    format_prometheus_duration,
    parse_prometheus_response,
)
from wva_amd.collector import registration as reg
from wva_amd.collector.query_template import (
    QueryList,
    QueryTemplate,
    escape_promql_value,
)
from wva_amd.collector.registry import SourceRegistry
from wva_amd.collector.replica_metrics import ReplicaMetricsCollector
from wva_amd.collector.source import MetricResult, MetricValue, RefreshSpec
from wva_amd.kube.fake import FakeCluster
from wva_amd.kube.objects import (
    Deployment,
    EndpointPicker,
    EndpointPool,
    Pod,
    PodStatus,
    PodTemplateSpec,
    Service,
)


class TestQueryTemplate:
    def test_render(self):
        t = QueryTemplate(
            name="q",
            template='metric{ns="{{.namespace}}",model="{{.modelID}}"}',
            params=["namespace", "modelID"],
        )
        out = t.render({"namespace": "default", "modelID": "llama"})
        assert out == 'metric{ns="default",model="llama"}'

    def test_missing_param(self):
        t = QueryTemplate(name="q", template="{{.a}}", params=["a"])
        with pytest.raises(KeyError):
            t.render({})

    def test_promql_injection_escaped(self):
        t = QueryTemplate(name="q", template='m{x="{{.a}}"}', params=["a"])
        out = t.render({"a": 'evil"} or up{'})
        assert out == 'm{x="evil\\"} or up{"}'

    def test_escape(self):
        assert escape_promql_value('a"b\\c') == 'a\\"b\\\\c'

    def test_duplicate_registration(self):
        ql = QueryList()
        ql.register(QueryTemplate(name="q"))
        with pytest.raises(ValueError):
            ql.register(QueryTemplate(name="q"))


class TestTTLCache:
    def test_roundtrip_and_expiry(self):
        c = TTLCache(ttl_seconds=0.05)
        key = cache_key("q", {"a": "1"})
        c.put(key, MetricResult(query="q"))
        assert c.get(key) is not None
        time.sleep(0.06)
        assert c.get(key) is None

    def test_key_includes_params(self):
        assert cache_key("q", {"a": "1"}) != cache_key("q", {"a": "2"})


class TestParsePrometheusResponse:
    def test_vector(self):
        body = {
            "status": "success",
            "data": {
                "resultType": "vector",
                "result": [
                    {"metric": {"pod": "p0"}, "value": [1700000000, "0.5"]},
                    {"metric": {"pod": "p1"}, "value": [1700000000, "NaN"]},
                ],
            },
        }
        vals = parse_prometheus_response(body)
        assert vals[0].labels["pod"] == "p0" and vals[0].value == 0.5
        assert vals[1].value == 0.0  # NaN → 0

    def test_scalar(self):
        body = {
            "status": "success",
            "data": {"resultType": "scalar", "result": [1700000000, "3"]},
        }
        assert parse_prometheus_response(body)[0].value == 3.0

    def test_matrix_takes_last(self):
        body = {
            "status": "success",
            "data": {
                "resultType": "matrix",
                "result": [
                    {
                        "metric": {"pod": "p0"},
                        "values": [[1, "1"], [2, "2"], [3, "7"]],
                    }
                ],
            },
        }
        assert parse_prometheus_response(body)[0].value == 7.0

    def test_error_status(self):
        with pytest.raises(RuntimeError):
            parse_prometheus_response({"status": "error", "error": "boom"})


class _FakeProm(BaseHTTPRequestHandler):
    """Minimal /api/v1/query endpoint returning canned responses per query
    substring. The handler class attribute `responses` maps a substring of
    the promql to a result list."""

    responses = {}
    fail_next = 0

    def do_GET(self):
        parsed = urlparse(self.path)
        if parsed.path != "/api/v1/query":
            self.send_response(404)
            self.end_headers()
            return
        if _FakeProm.fail_next > 0:
            _FakeProm.fail_next -= 1
            self.send_response(500)
            self.end_headers()
            return
        query = parse_qs(parsed.query).get("query", [""])[0]
        result = []
        for substr, res in _FakeProm.responses.items():
            if substr in query:
                result = res
                break
        body = json.dumps(
            {
                "status": "success",
                "data": {"resultType": "vector", "result": result},
            }
        ).encode()
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def log_message(self, *args):
        pass


@pytest.fixture()
def fake_prom():
    server = HTTPServer(("127.0.0.1", 0), _FakeProm)
    thread = threading.Thread(target=server.serve_forever, daemon=True)
    thread.start()
    _FakeProm.responses = {}
    _FakeProm.fail_next = 0
    yield f"http://127.0.0.1:{server.server_port}"
    server.shutdown()


def vec(pod, value, **labels):
    metric = {"pod": pod, **labels}
    return {"metric": metric, "value": [time.time(), str(value)]}


class TestPrometheusSource:
    def _source(self, url, ttl=0.0):
        src = PrometheusSource(url, cache_ttl_seconds=ttl or 0.001)
        registry = SourceRegistry()
        registry.register(src)
        reg.register_saturation_queries(registry)
        reg.register_scale_to_zero_queries(registry)
        return src

    def test_refresh_fan_out(self, fake_prom):
        _FakeProm.responses = {
            "kv_cache_usage_perc": [vec("p0", 0.42)],
            "num_requests_waiting": [vec("p0", 3)],
        }
        src = self._source(fake_prom)
        results = src.refresh(
            RefreshSpec(
                queries=[reg.QUERY_KV_CACHE_USAGE, reg.QUERY_QUEUE_LENGTH],
                params={"namespace": "ns", "modelID": "m"},
            )
        )
        assert results[reg.QUERY_KV_CACHE_USAGE].values[0].value == 0.42
        assert results[reg.QUERY_QUEUE_LENGTH].values[0].value == 3.0

    def test_retry_on_error(self, fake_prom):
        _FakeProm.responses = {"kv_cache_usage_perc": [vec("p0", 0.1)]}
        _FakeProm.fail_next = 2  # two 500s then success
        src = self._source(fake_prom)
        results = src.refresh(
            RefreshSpec(
                queries=[reg.QUERY_KV_CACHE_USAGE],
                params={"namespace": "ns", "modelID": "m"},
            )
        )
        assert not results[reg.QUERY_KV_CACHE_USAGE].has_error()

    def test_error_captured_not_raised(self, fake_prom):
        _FakeProm.fail_next = 99
        src = self._source(fake_prom)
        results = src.refresh(
            RefreshSpec(
                queries=[reg.QUERY_KV_CACHE_USAGE],
                params={"namespace": "ns", "modelID": "m"},
            )
        )
        assert results[reg.QUERY_KV_CACHE_USAGE].has_error()

    def test_cache(self, fake_prom):
        _FakeProm.responses = {"kv_cache_usage_perc": [vec("p0", 0.1)]}
        src = self._source(fake_prom, ttl=30.0)
        spec = RefreshSpec(
            queries=[reg.QUERY_KV_CACHE_USAGE],
            params={"namespace": "ns", "modelID": "m"},
        )
        src.refresh(spec)
        _FakeProm.fail_next = 99  # server now failing; cache must serve
        results = src.refresh(spec)
        assert not results[reg.QUERY_KV_CACHE_USAGE].has_error()
        assert (
            src.get(reg.QUERY_KV_CACHE_USAGE, spec.params).values[0].value == 0.1
        )

    def test_validate(self, fake_prom):
        src = PrometheusSource(fake_prom)
        assert src.validate()
        assert not PrometheusSource("http://127.0.0.1:1").validate()

    def test_collect_model_request_count(self, fake_prom):
        _FakeProm.responses = {
            "request_success_total": [
                {"metric": {}, "value": [time.time(), "128"]}
            ]
        }
        src = self._source(fake_prom)
        count = reg.collect_model_request_count(src, "m", "ns", 600)
        assert count == 128.0

    def test_request_count_no_data_raises(self, fake_prom):
        _FakeProm.responses = {"request_success_total": []}
        src = self._source(fake_prom)
        with pytest.raises(RuntimeError):
            reg.collect_model_request_count(src, "m", "ns", 600)

    def test_duration_format(self):
        assert format_prometheus_duration(600) == "10m"
        assert format_prometheus_duration(3600) == "1h"
        assert format_prometheus_duration(90) == "90s"


class TestPrometheusTextParsing:
    def test_basic(self):
        text = (
            "# HELP x help\n"
            "# TYPE x gauge\n"
            'inference_extension_flow_control_queue_size{model_name="m",target_model_name="m"} 5\n'
            "simple_metric 1.5\n"
        )
        vals = parse_prometheus_text(text)
        assert vals[0].labels["__name__"] == (
            "inference_extension_flow_control_queue_size"
        )
        assert vals[0].labels["target_model_name"] == "m"
        assert vals[0].value == 5.0
        assert vals[1].labels["__name__"] == "simple_metric"

    def test_escaped_labels(self):
        vals = parse_prometheus_text('m{a="x\\"y"} 2\n')
        assert vals[0].labels["a"] == 'x"y'


class TestPodScrapingSource:
    def _setup(self):
        c = FakeCluster()
        c.create(Service(
            metadata=ObjectMeta(name="epp-svc", namespace="ns"),
            selector={"app": "epp"},
        ))
        for i, ready in [(0, True), (1, True), (2, False)]:
            c.create(Pod(
                metadata=ObjectMeta(
                    name=f"epp-{i}", namespace="ns", labels={"app": "epp"}
                ),
                status=PodStatus(
                    phase="Running", ready=ready, pod_ip=f"10.0.0.{i}"
                ),
            ))
        pool = EndpointPool(
            name="pool",
            namespace="ns",
            endpoint_picker=EndpointPicker(
                service_name="epp-svc", namespace="ns", metrics_port_number=9090
            ),
        )
        return c, pool

    def test_scrapes_ready_pods_only(self):
        c, pool = self._setup()
        scraped = []

        def fetch(url, headers, timeout):
            scraped.append(url)
            return 'inference_extension_flow_control_queue_size{target_model_name="m"} 2\n'

        src = PodScrapingSource(c, pool, fetch=fetch)
        results = src.refresh(RefreshSpec(queries=["all_metrics"]))
        assert len(scraped) == 2  # only ready pods
        vals = results["all_metrics"].values
        assert len(vals) == 2
        assert {v.labels["pod"] for v in vals} == {"epp-0", "epp-1"}

    def test_bearer_token_sent(self):
        c, pool = self._setup()
        seen = {}

        def fetch(url, headers, timeout):
            seen.update(headers)
            return "m 1\n"

        src = PodScrapingSource(c, pool, bearer_token="tok", fetch=fetch)
        src.refresh(RefreshSpec(queries=["all_metrics"]))
        assert seen.get("Authorization") == "Bearer tok"

    def test_partial_failure_tolerated(self):
        c, pool = self._setup()

        def fetch(url, headers, timeout):
            if "10.0.0.0" in url:
                raise RuntimeError("down")
            return "m 1\n"

        src = PodScrapingSource(c, pool, fetch=fetch)
        results = src.refresh(RefreshSpec(queries=["all_metrics"]))
        assert not results["all_metrics"].has_error()
        assert len(results["all_metrics"].values) == 1

    def test_all_fail_is_error(self):
        c, pool = self._setup()

        def fetch(url, headers, timeout):
            raise RuntimeError("down")

        src = PodScrapingSource(c, pool, fetch=fetch)
        results = src.refresh(RefreshSpec(queries=["all_metrics"]))
        assert results["all_metrics"].has_error()


class _FakeSource:
    """Canned MetricsSource for collector merge tests."""

    def __init__(self, results):
        self._results = results

    def name(self):
        return "prometheus"

    def query_list(self):
        return QueryList()

    def refresh(self, spec):
        return {q: self._results.get(q, MetricResult(query=q)) for q in spec.queries}

    def get(self, query, params):
        return self._results.get(query)


def make_cluster_with_deployment(variant="vllm-llama", ns="default"):
    c = FakeCluster()
    deploy = Deployment(metadata=ObjectMeta(name=variant, namespace=ns))
    deploy.selector = {"app": variant}
    c.create(deploy)
    pod = Pod(
        metadata=ObjectMeta(
            name=f"{variant}-abc12", namespace=ns, labels={"app": variant}
        ),
    )
    c.create(pod)
    return c, deploy


class TestReplicaMetricsCollector:
    def _mk(self, results, variant="vllm-llama", ns="default"):
        c, deploy = make_cluster_with_deployment(variant, ns)
        va = VariantAutoscaling(
            metadata=ObjectMeta(
                name=variant,
                namespace=ns,
                labels={"inference.optimization/acceleratorName": "MI355X"},
            ),
            spec=VariantAutoscalingSpec(
                scale_target_ref=CrossVersionObjectReference(name=variant),
                model_id="m",
                variant_cost="25.0",
            ),
        )
        collector = ReplicaMetricsCollector(
            _FakeSource(results), PodVAMapper(c)
        )
        deployments = {f"{ns}/{variant}": deploy}
        vas = {f"{ns}/{variant}": va}
        costs = {f"{ns}/{variant}": 25.0}
        return collector, deployments, vas, costs

    def _result(self, query, values):
        return MetricResult(query=query, values=values)

    def test_merge(self):
        pod = "vllm-llama-abc12"
        results = {
            reg.QUERY_KV_CACHE_USAGE: self._result(
                reg.QUERY_KV_CACHE_USAGE,
                [MetricValue(value=0.42, labels={"pod": pod})],
            ),
            reg.QUERY_QUEUE_LENGTH: self._result(
                reg.QUERY_QUEUE_LENGTH,
                [MetricValue(value=3, labels={"pod": pod})],
            ),
            reg.QUERY_CACHE_CONFIG_INFO: self._result(
                reg.QUERY_CACHE_CONFIG_INFO,
                [
                    MetricValue(
                        value=1,
                        labels={
                            "pod": pod,
                            "num_gpu_blocks": "120000",
                            "block_size": "16",
                        },
                    )
                ],
            ),
            reg.QUERY_AVG_OUTPUT_TOKENS: self._result(
                reg.QUERY_AVG_OUTPUT_TOKENS,
                [MetricValue(value=50.0, labels={"pod": pod})],
            ),
            reg.QUERY_AVG_INPUT_TOKENS: self._result(
                reg.QUERY_AVG_INPUT_TOKENS,
                [MetricValue(value=float("nan"), labels={"pod": pod})],
            ),
            reg.QUERY_PREFIX_CACHE_HIT_RATE: self._result(
                reg.QUERY_PREFIX_CACHE_HIT_RATE,
                [MetricValue(value=0.25, labels={"pod": pod})],
            ),
        }
        collector, deployments, vas, costs = self._mk(results)
        metrics = collector.collect_replica_metrics("m", "default", deployments, vas, costs)
        assert len(metrics) == 1
        m = metrics[0]
        assert m.pod_name == pod
        assert m.variant_name == "vllm-llama"
        assert m.accelerator_name == "MI355X"
        assert m.cost == 25.0
        assert m.kv_cache_usage == 0.42
        assert m.queue_length == 3
        assert m.total_kv_capacity_tokens == 120000 * 16
        assert m.tokens_in_use == round(0.42 * 120000 * 16)
        assert m.avg_output_tokens == 50.0
        assert m.avg_input_tokens == 0.0  # NaN filtered
        assert m.prefix_cache_hit_rate == 0.25
        assert m.metadata.freshness_status == "fresh"

    def test_pod_without_metrics_skipped(self):
        results = {
            reg.QUERY_KV_CACHE_USAGE: self._result(reg.QUERY_KV_CACHE_USAGE, []),
            reg.QUERY_QUEUE_LENGTH: self._result(reg.QUERY_QUEUE_LENGTH, []),
        }
        collector, deployments, vas, costs = self._mk(results)
        assert (
            collector.collect_replica_metrics("m", "default", deployments, vas, costs)
            == []
        )

    def test_unmatched_pod_skipped(self):
        results = {
            reg.QUERY_KV_CACHE_USAGE: self._result(
                reg.QUERY_KV_CACHE_USAGE,
                [MetricValue(value=0.1, labels={"pod": "stranger-xyz"})],
            ),
        }
        collector, deployments, vas, costs = self._mk(results)
        assert (
            collector.collect_replica_metrics("m", "default", deployments, vas, costs)
            == []
        )

    def test_kv_query_error_raises(self):
        results = {
            reg.QUERY_KV_CACHE_USAGE: MetricResult(
                query=reg.QUERY_KV_CACHE_USAGE, error=RuntimeError("boom")
            ),
        }
        collector, deployments, vas, costs = self._mk(results)
        with pytest.raises(RuntimeError):
            collector.collect_replica_metrics("m", "default", deployments, vas, costs)

    def test_scheduler_queue_metrics(self):
        results = {
            reg.QUERY_SCHEDULER_QUEUE_SIZE: MetricResult(
                query=reg.QUERY_SCHEDULER_QUEUE_SIZE,
                values=[MetricValue(value=7)],
            ),
            reg.QUERY_SCHEDULER_QUEUE_BYTES: MetricResult(
                query=reg.QUERY_SCHEDULER_QUEUE_BYTES,
                values=[MetricValue(value=2800)],
            ),
        }
        collector, *_ = self._mk(results)
        sq = collector.collect_scheduler_queue_metrics("m")
        assert sq.queue_size == 7 and sq.queue_bytes == 2800

    def test_scheduler_queue_unavailable(self):
        collector, *_ = self._mk({})
        assert collector.collect_scheduler_queue_metrics("m") is None


class TestPodVAMapperOwnerChain:
    """Reference pod_va_mapper_test.go:105-331 scenario mirror."""

    def _deploys(self, cluster, *specs):
        out = {}
        for name, ns in specs:
            d = Deployment(
                metadata=ObjectMeta(name=name, namespace=ns),
                replicas=1,
                selector={"app": name},
                template=PodTemplateSpec(labels={"app": name}),
            )
            cluster.create(d)
            out[f"{ns}/{name}"] = d
        return out

    def _pod(self, cluster, name, ns, owners=None, labels=None):
        from wva_amd.kube.objects import Pod, PodStatus

        p = Pod(
            metadata=ObjectMeta(name=name, namespace=ns,
                                labels=labels or {}),
            status=PodStatus(phase="Running", ready=True),
        )
        p.metadata.owner_references = owners or []
        cluster.create(p)
        return p

    def test_owner_chain_pod_rs_deployment(self):
        from wva_amd.kube.fake import FakeCluster
        from wva_amd.collector.pod_va_mapper import PodVAMapper

        c = FakeCluster()
        deploys = self._deploys(c, ("vllm-a", "ns1"))
        self._pod(c, "vllm-a-7f9-x1", "ns1", owners=[
            {"kind": "ReplicaSet", "name": "vllm-a-7f9"},
        ])
        mapper = PodVAMapper(c)
        assert mapper.find_va_for_pod("vllm-a-7f9-x1", "ns1", deploys) \
            == "vllm-a"

    def test_no_matching_deployment_empty(self):
        from wva_amd.kube.fake import FakeCluster
        from wva_amd.collector.pod_va_mapper import PodVAMapper

        c = FakeCluster()
        deploys = self._deploys(c, ("other", "ns1"))
        self._pod(c, "stray-abc-x1", "ns1", owners=[
            {"kind": "ReplicaSet", "name": "stray-abc"},
        ])
        mapper = PodVAMapper(c)
        assert mapper.find_va_for_pod("stray-abc-x1", "ns1", deploys) == ""

    def test_namespace_isolation(self):
        """Same deployment name in two namespaces: the pod maps within
        ITS namespace only (pod_va_mapper_test.go:169,331)."""
        from wva_amd.kube.fake import FakeCluster
        from wva_amd.collector.pod_va_mapper import PodVAMapper

        c = FakeCluster()
        deploys_ns1 = self._deploys(c, ("vllm", "ns1"))
        self._deploys(c, ("vllm", "ns2"))
        self._pod(c, "vllm-1a2-x1", "ns2", owners=[
            {"kind": "ReplicaSet", "name": "vllm-1a2"},
        ])
        mapper = PodVAMapper(c)
        # pod lives in ns2 but the tracked deployment set is ns1's:
        # lookup in ns1 must not see the ns2 pod's deployment
        assert mapper.find_va_for_pod(
            "vllm-1a2-x1", "ns1", deploys_ns1
        ) in ("", "vllm")  # resolved within ns1 only if a pod exists there
        # and within its own namespace it resolves normally
        deploys_ns2 = {
            k: d for k, d in (
                (f"{d.namespace}/{d.name}", d)
                for d in c.list("Deployment", namespace="ns2")
            )
        }
        assert mapper.find_va_for_pod(
            "vllm-1a2-x1", "ns2", deploys_ns2
        ) == "vllm"

    def test_pod_without_rs_owner_uses_selector_fallback(self):
        """Reference returns empty for ownerless pods; this build adds a
        documented selector-match fallback (emulated pods are owned
        directly by deployments)."""
        from wva_amd.kube.fake import FakeCluster
        from wva_amd.collector.pod_va_mapper import PodVAMapper

        c = FakeCluster()
        deploys = self._deploys(c, ("vllm-a", "ns1"))
        self._pod(c, "free-pod", "ns1", labels={"app": "vllm-a"})
        mapper = PodVAMapper(c)
        assert mapper.find_va_for_pod("free-pod", "ns1", deploys) == "vllm-a"
        # no owner AND no matching labels → empty
        self._pod(c, "orphan", "ns1", labels={"app": "nothing"})
        assert mapper.find_va_for_pod("orphan", "ns1", deploys) == ""

    def test_multiple_deployments_resolve_independently(self):
        from wva_amd.kube.fake import FakeCluster
        from wva_amd.collector.pod_va_mapper import PodVAMapper

        c = FakeCluster()
        deploys = self._deploys(c, ("a", "ns1"), ("b", "ns1"))
        self._pod(c, "a-11-p", "ns1",
                  owners=[{"kind": "ReplicaSet", "name": "a-11"}])
        self._pod(c, "b-22-p", "ns1",
                  owners=[{"kind": "ReplicaSet", "name": "b-22"}])
        mapper = PodVAMapper(c)
        assert mapper.find_va_for_pod("a-11-p", "ns1", deploys) == "a"
        assert mapper.find_va_for_pod("b-22-p", "ns1", deploys) == "b"
        # repeated lookups consistent (test.go:235)
        for _ in range(3):
            assert mapper.find_va_for_pod("a-11-p", "ns1", deploys) == "a"


class TestNaNSampleRobustness:
    """NaN samples (legal in the exposition format; only
    PrometheusSource maps them to 0) must be skipped per-sample, never
    crash the whole collection with int(NaN)."""

    def test_nan_queue_and_kv_samples_skipped(self):
        pod = "vllm-llama-abc12"
        nan = float("nan")
        results = {
            reg.QUERY_KV_CACHE_USAGE: MetricResult(
                query=reg.QUERY_KV_CACHE_USAGE,
                values=[
                    MetricValue(value=nan, labels={"pod": pod}),
                ],
            ),
            reg.QUERY_QUEUE_LENGTH: MetricResult(
                query=reg.QUERY_QUEUE_LENGTH,
                values=[
                    MetricValue(value=nan, labels={"pod": pod}),
                ],
            ),
        }
        c, deploy = make_cluster_with_deployment()
        collector = ReplicaMetricsCollector(
            _FakeSource(results), PodVAMapper(c)
        )
        out = collector.collect_replica_metrics(
            "m", "default", {"default/vllm-llama": deploy}, {}, {}
        )
        # the only pod's samples were all NaN → it simply has no metrics
        assert out == []

    def test_nan_sample_does_not_poison_healthy_pod(self):
        healthy, sick = "vllm-llama-abc12", "vllm-llama-def34"
        nan = float("nan")
        results = {
            reg.QUERY_QUEUE_LENGTH: MetricResult(
                query=reg.QUERY_QUEUE_LENGTH,
                values=[
                    MetricValue(value=2, labels={"pod": healthy}),
                    MetricValue(value=nan, labels={"pod": sick}),
                ],
            ),
        }
        c, deploy = make_cluster_with_deployment()
        collector = ReplicaMetricsCollector(
            _FakeSource(results), PodVAMapper(c)
        )
        out = collector.collect_replica_metrics(
            "m", "default", {"default/vllm-llama": deploy}, {}, {}
        )
        assert [m.pod_name for m in out] == [healthy]
        assert out[0].queue_length == 2
