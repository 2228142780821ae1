"""Runtime + controller tests: executor retry, manager dispatch + leader
election, datastore, indexers, predicates, ConfigMap reconciler, metrics
emitter, config loader. Mirrors the reference's envtest/unit coverage for
cmd/main wiring, internal/controller and internal/config.
"""
import os
import threading
import time

import pytest
from prometheus_client import CollectorRegistry

from wva_amd.api.types import (
    CrossVersionObjectReference,
    ObjectMeta,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.config.config import Config
from wva_amd.config.loader import load_config
from wva_amd.config.validation import ConfigLoadError
from wva_amd.controllers.configmap import (
    ConfigMapReconciler,
    parse_saturation_configmap,
)
from wva_amd.controllers.predicates import (
    configmap_predicate,
    deployment_predicate,
    variant_autoscaling_predicate,
)
from wva_amd.datastore.datastore import Datastore
from wva_amd.engines.common import DecisionCache, DecisionTrigger
from wva_amd.kube.fake import ADDED, DELETED, MODIFIED, FakeCluster, WatchEvent
from wva_amd.kube.indexers import VAIndex
from wva_amd.kube.objects import ConfigMap, Deployment, EndpointPool
from wva_amd.metrics.metrics import MetricsEmitter
from wva_amd.runtime.executor import PollingExecutor
from wva_amd.runtime.manager import LeaderElector, Manager


def make_va(name="va1", ns="default", target=None, labels=None):
    return VariantAutoscaling(
        metadata=ObjectMeta(name=name, namespace=ns, labels=labels or {}),
        spec=VariantAutoscalingSpec(
            scale_target_ref=CrossVersionObjectReference(name=target or name),
            model_id="m",
        ),
    )


class TestPollingExecutor:
    def test_tick_retry_until_success(self):
        calls = []

        def tick():
            calls.append(1)
            if len(calls) < 3:
                raise RuntimeError("boom")

        ex = PollingExecutor(999, tick, name="t")
        ex.INITIAL_RETRY_BACKOFF_SECONDS = 0.01
        # run_once retries internally with backoff until success
        import wva_amd.runtime.executor as exmod

        old = exmod.INITIAL_RETRY_BACKOFF_SECONDS
        exmod.INITIAL_RETRY_BACKOFF_SECONDS = 0.01
        try:
            ex.run_once()
        finally:
            exmod.INITIAL_RETRY_BACKOFF_SECONDS = old
        assert len(calls) == 3

    def test_periodic_loop(self):
        calls = []
        ex = PollingExecutor(0.02, lambda: calls.append(1), name="t")
        ex.start()
        time.sleep(0.15)
        ex.stop()
        assert len(calls) >= 3

    def test_stop_interrupts_backoff(self):
        ex = PollingExecutor(10, lambda: (_ for _ in ()).throw(RuntimeError()), name="t")
        ex.start()
        time.sleep(0.05)
        t0 = time.time()
        ex.stop()
        assert time.time() - t0 < 3.0


class TestLeaderElection:
    def test_acquire_and_renew(self):
        c = FakeCluster()
        e1 = LeaderElector(c, "test-lease", identity="a", lease_duration=1.0)
        assert e1.try_acquire_or_renew()
        assert e1.try_acquire_or_renew()  # renew

    def test_second_elector_blocked_until_expiry(self):
        c = FakeCluster()
        e1 = LeaderElector(c, "test-lease", identity="a", lease_duration=0.1)
        e2 = LeaderElector(c, "test-lease", identity="b", lease_duration=0.1)
        assert e1.try_acquire_or_renew()
        assert not e2.try_acquire_or_renew()
        time.sleep(0.15)  # lease expires
        assert e2.try_acquire_or_renew()

    def test_release_on_cancel_fast_failover(self):
        c = FakeCluster()
        e1 = LeaderElector(c, "test-lease", identity="a", lease_duration=60.0)
        e2 = LeaderElector(c, "test-lease", identity="b", lease_duration=60.0)
        assert e1.try_acquire_or_renew()
        e1.release()
        assert e2.try_acquire_or_renew()  # no wait for 60s expiry


class TestManagerDispatch:
    def test_watch_event_drives_reconcile(self):
        c = FakeCluster()
        config = Config()
        config.mark_bootstrap_complete()
        mgr = Manager(c, config)
        seen = []
        mgr.register_reconciler(
            ["VariantAutoscaling"],
            lambda e: e.type == ADDED,
            lambda ns, name: seen.append((ns, name)),
        )
        mgr.start()
        try:
            c.create(make_va("v1"))
            deadline = time.time() + 2
            while not seen and time.time() < deadline:
                time.sleep(0.02)
            assert seen == [("default", "v1")]
        finally:
            mgr.stop()

    def test_decision_trigger_feeds_va_reconciler(self):
        c = FakeCluster()
        config = Config()
        config.mark_bootstrap_complete()
        trigger = DecisionTrigger()
        mgr = Manager(c, config, trigger)
        seen = []
        mgr.register_reconciler(
            ["VariantAutoscaling"],
            lambda e: False,
            lambda ns, name: seen.append((ns, name)),
            is_va_reconciler=True,
        )
        mgr.start()
        try:
            trigger.push("ns1", "va9")
            deadline = time.time() + 2
            while not seen and time.time() < deadline:
                time.sleep(0.02)
            assert seen == [("ns1", "va9")]
        finally:
            mgr.stop()

    def test_leader_gates_runnables(self):
        c = FakeCluster()
        config = Config()
        config.infra.enable_leader_election = True
        config.infra.retry_period_seconds = 0.05
        config.mark_bootstrap_complete()
        started = threading.Event()

        class R:
            def start(self):
                started.set()

            def stop(self):
                pass

        mgr = Manager(c, config)
        mgr.add_runnable(R())
        mgr.start()
        try:
            assert started.wait(timeout=2.0)
            assert mgr.is_leader()
        finally:
            mgr.stop()


class TestDatastore:
    def test_pool_registry(self):
        c = FakeCluster()
        ds = Datastore(c)
        ds.pool_set(EndpointPool(name="p", namespace="ns", selector={"a": "b"}))
        assert ds.pool_get("ns", "p") is not None
        assert ds.pool_source("ns", "p") is not None
        assert ds.pool_get_from_labels("ns", {"a": "b", "x": "y"}).name == "p"
        assert ds.pool_get_from_labels("ns", {"a": "wrong"}) is None
        assert ds.pool_get_from_labels("other", {"a": "b"}) is None
        ds.pool_delete("ns", "p")
        assert ds.pool_get("ns", "p") is None

    def test_namespace_tracking(self):
        ds = Datastore(FakeCluster())
        ds.namespace_track("a")
        assert ds.namespace_is_tracked("a")
        assert not ds.namespace_is_tracked("b")
        ds.namespace_untrack("a")
        assert not ds.namespace_is_tracked("a")


class TestIndexers:
    def test_find_va_for_deployment(self):
        c = FakeCluster()
        c.create(make_va("va1", target="deploy1"))
        idx = VAIndex(c)
        va = idx.find_va_for_deployment("default", "deploy1")
        assert va is not None and va.name == "va1"
        assert idx.find_va_for_deployment("default", "other") is None

    def test_duplicate_target_rejected(self):
        c = FakeCluster()
        c.create(make_va("va1", target="deploy1"))
        c.create(make_va("va2", target="deploy1"))
        idx = VAIndex(c)
        with pytest.raises(ValueError):
            idx.find_va_for_deployment("default", "deploy1")
        assert idx.validate_unique_targets()


class TestPredicates:
    def test_va_predicate_create_only(self):
        c = FakeCluster()
        pred = variant_autoscaling_predicate(c)
        va = make_va()
        assert pred(WatchEvent(type=ADDED, kind="VariantAutoscaling", obj=va))
        assert not pred(WatchEvent(type=MODIFIED, kind="VariantAutoscaling", obj=va))
        assert pred(WatchEvent(type=DELETED, kind="VariantAutoscaling", obj=va))

    def test_controller_instance_isolation(self):
        c = FakeCluster()
        pred = variant_autoscaling_predicate(c)
        labeled = make_va(labels={"wva.llmd.ai/controller-instance": "other"})
        assert not pred(WatchEvent(type=ADDED, kind="VariantAutoscaling", obj=labeled))
        os.environ["CONTROLLER_INSTANCE"] = "other"
        try:
            assert pred(WatchEvent(type=ADDED, kind="VariantAutoscaling", obj=labeled))
            assert not pred(WatchEvent(type=ADDED, kind="VariantAutoscaling", obj=make_va()))
        finally:
            del os.environ["CONTROLLER_INSTANCE"]

    def test_deployment_predicate(self):
        pred = deployment_predicate()
        d = Deployment(metadata=ObjectMeta(name="d", namespace="ns"))
        assert pred(WatchEvent(type=ADDED, kind="Deployment", obj=d))
        assert not pred(WatchEvent(type=MODIFIED, kind="Deployment", obj=d))
        assert pred(WatchEvent(type=DELETED, kind="Deployment", obj=d))

    def test_configmap_predicate_well_known_names(self):
        pred = configmap_predicate()
        good = ConfigMap(metadata=ObjectMeta(
            name="wva-saturation-scaling-config", namespace="ns"))
        bad = ConfigMap(metadata=ObjectMeta(name="random", namespace="ns"))
        assert pred(WatchEvent(type=MODIFIED, kind="ConfigMap", obj=good))
        assert not pred(WatchEvent(type=MODIFIED, kind="ConfigMap", obj=bad))


class TestConfigMapReconciler:
    def _setup(self, controller_ns="wva-system"):
        os.environ["POD_NAMESPACE"] = controller_ns
        c = FakeCluster()
        config = Config()
        ds = Datastore(c)
        rec = ConfigMapReconciler(c, config, ds)
        return c, config, ds, rec

    def test_global_saturation_config(self):
        c, config, ds, rec = self._setup()
        c.create(ConfigMap(
            metadata=ObjectMeta(
                name="wva-saturation-scaling-config", namespace="wva-system"
            ),
            data={"default": "kvCacheThreshold: 0.7\nqueueLengthThreshold: 9\n"},
        ))
        rec.reconcile("wva-system", "wva-saturation-scaling-config")
        cfg = config.saturation_config()
        assert cfg.kv_cache_threshold == 0.7
        assert cfg.queue_length_threshold == 9

    def test_namespace_local_override(self):
        c, config, ds, rec = self._setup()
        ds.namespace_track("team-a")
        c.create(ConfigMap(
            metadata=ObjectMeta(
                name="wva-saturation-scaling-config", namespace="team-a"
            ),
            data={"default": "kvCacheThreshold: 0.5\n"},
        ))
        rec.reconcile("team-a", "wva-saturation-scaling-config")
        assert config.saturation_config_for_namespace("team-a").kv_cache_threshold == 0.5
        assert config.saturation_config_for_namespace("other").kv_cache_threshold == 0.8

    def test_untracked_namespace_ignored(self):
        c, config, ds, rec = self._setup()
        c.create(ConfigMap(
            metadata=ObjectMeta(
                name="wva-saturation-scaling-config", namespace="stranger"
            ),
            data={"default": "kvCacheThreshold: 0.5\n"},
        ))
        rec.reconcile("stranger", "wva-saturation-scaling-config")
        assert config.saturation_config_for_namespace("stranger").kv_cache_threshold == 0.8

    def test_deletion_removes_override(self):
        c, config, ds, rec = self._setup()
        ds.namespace_track("team-a")
        c.create(ConfigMap(
            metadata=ObjectMeta(
                name="wva-saturation-scaling-config", namespace="team-a"
            ),
            data={"default": "kvCacheThreshold: 0.5\n"},
        ))
        rec.reconcile("team-a", "wva-saturation-scaling-config")
        c.delete("ConfigMap", "team-a", "wva-saturation-scaling-config")
        rec.reconcile("team-a", "wva-saturation-scaling-config")
        assert config.saturation_config_for_namespace("team-a").kv_cache_threshold == 0.8

    def test_invalid_config_keeps_previous(self):
        c, config, ds, rec = self._setup()
        cm = ConfigMap(
            metadata=ObjectMeta(
                name="wva-saturation-scaling-config", namespace="wva-system"
            ),
            data={"default": "kvCacheThreshold: 0.7\n"},
        )
        c.create(cm)
        rec.reconcile("wva-system", "wva-saturation-scaling-config")
        cm.data = {"default": "kvCacheThreshold: 7.0\n"}  # invalid > 1
        c.update(cm)
        rec.reconcile("wva-system", "wva-saturation-scaling-config")
        assert config.saturation_config().kv_cache_threshold == 0.7

    def test_per_model_overrides(self):
        parsed = parse_saturation_configmap({
            "default": "kvCacheThreshold: 0.8\n",
            "llama": "model_id: m1\nnamespace: ns\nkvCacheThreshold: 0.6\n",
        })
        assert parsed.for_model("m1", "ns").kv_cache_threshold == 0.6
        assert parsed.for_model("other", "ns").kv_cache_threshold == 0.8

    def test_bootstrap_marks_ready(self):
        c, config, ds, rec = self._setup()
        assert not config.is_bootstrap_complete()
        rec.bootstrap_initial_configmaps()
        assert config.is_bootstrap_complete()


class TestMetricsEmitter:
    def test_labels_and_ratio(self):
        reg = CollectorRegistry()
        e = MetricsEmitter(registry=reg, controller_instance="")
        e.emit_replica_metrics("v", "ns", current=2, desired=4, accelerator_type="MI355X")
        samples = {}
        for fam in reg.collect():
            for s in fam.samples:
                samples[s.name] = s
        assert samples["wva_desired_replicas"].value == 4
        assert samples["wva_current_replicas"].value == 2
        assert samples["wva_desired_ratio"].value == 2.0
        assert samples["wva_desired_replicas"].labels == {
            "variant_name": "v", "namespace": "ns", "accelerator_type": "MI355X"
        }

    def test_zero_current_ratio_is_desired(self):
        reg = CollectorRegistry()
        e = MetricsEmitter(registry=reg, controller_instance="")
        e.emit_replica_metrics("v", "ns", current=0, desired=3, accelerator_type="A")
        for fam in reg.collect():
            if fam.name == "wva_desired_ratio":
                assert fam.samples[0].value == 3.0

    def test_scaling_counter(self):
        reg = CollectorRegistry()
        e = MetricsEmitter(registry=reg, controller_instance="c1")
        e.emit_replica_scaling_metrics("v", "ns", "up", "saturation")
        for fam in reg.collect():
            if fam.name == "wva_replica_scaling":  # counter family name
                s = fam.samples[0]
                assert s.labels["direction"] == "up"
                assert s.labels["controller_instance"] == "c1"


class TestConfigLoader:
    def test_defaults(self):
        cfg = load_config(env={}, require_prometheus=False)
        assert cfg.infra.optimization_interval_seconds == 60.0
        assert cfg.infra.lease_duration_seconds == 60.0
        assert cfg.scale_from_zero_max_concurrency() == 10
        assert cfg.cache.ttl_seconds == 30.0

    def test_env_overrides_defaults(self):
        cfg = load_config(
            env={"GLOBAL_OPT_INTERVAL": "30s", "WVA_SCALE_TO_ZERO": "true"},
            require_prometheus=False,
        )
        assert cfg.infra.optimization_interval_seconds == 30.0
        assert cfg.scale_to_zero_enabled()

    def test_flags_override_env(self):
        cfg = load_config(
            flags={"V": 4},
            env={"V": "1"},
            require_prometheus=False,
        )
        assert cfg.infra.logger_verbosity == 4

    def test_prometheus_required(self):
        with pytest.raises(ConfigLoadError):
            load_config(env={}, require_prometheus=True)
        cfg = load_config(
            env={"PROMETHEUS_BASE_URL": "http://prom:9090"},
            require_prometheus=True,
        )
        assert cfg.prometheus.base_url == "http://prom:9090"

    def test_invalid_leader_timings(self):
        with pytest.raises(ConfigLoadError):
            load_config(
                env={
                    "LEADER_ELECTION_LEASE_DURATION": "10s",
                    "LEADER_ELECTION_RENEW_DEADLINE": "50s",
                },
                require_prometheus=False,
            )

    def test_invalid_bind_address_rejected(self):
        with pytest.raises(ConfigLoadError):
            load_config(env={"METRICS_BIND_ADDRESS": "not a bind addr"},
                        require_prometheus=False)

    def test_prometheus_url_scheme_enforced(self):
        with pytest.raises(ConfigLoadError):
            load_config(env={"PROMETHEUS_BASE_URL": "ftp://prom:9090"})
        with pytest.raises(ConfigLoadError):
            load_config(env={})  # required when prometheus is required
        cfg = load_config(env={"PROMETHEUS_BASE_URL": "https://prom:9090"})
        assert cfg.prometheus.base_url == "https://prom:9090"

    def test_nonpositive_intervals_rejected(self):
        with pytest.raises(ConfigLoadError):
            load_config(env={"GLOBAL_OPT_INTERVAL": "0s"},
                        require_prometheus=False)
        with pytest.raises(ConfigLoadError):
            load_config(env={"PROMETHEUS_METRICS_CACHE_TTL": "-5s",
                             "PROMETHEUS_BASE_URL": "http://p:9090"})

    def test_all_errors_aggregated_in_one_raise(self):
        """Fail-fast startup reports EVERY problem at once
        (validation.go aggregation), not just the first."""
        with pytest.raises(ConfigLoadError) as ei:
            load_config(env={
                "METRICS_BIND_ADDRESS": "bogus addr",
                "GLOBAL_OPT_INTERVAL": "0s",
            }, require_prometheus=True)
        msg = str(ei.value)
        assert "METRICS_BIND_ADDRESS" in msg
        assert "GLOBAL_OPT_INTERVAL" in msg
        assert "PROMETHEUS_BASE_URL" in msg


class TestInferencePoolReconciler:
    """Reference inferencepool_reconciler.go:41-118 + pool.go:34-148:
    InferencePool → EndpointPool conversion, metrics-port-by-name, and
    deletion-driven pool removal."""

    def _stack(self):
        from wva_amd.controllers.inferencepool import InferencePoolReconciler
        from wva_amd.datastore.datastore import Datastore

        cluster = FakeCluster()
        ds = Datastore(cluster)
        return cluster, ds, InferencePoolReconciler(cluster, ds)

    def test_conversion_and_metrics_port_by_name(self):
        from wva_amd.kube.objects import (
            InferencePool, Service, ServicePort,
        )

        cluster, ds, rec = self._stack()
        cluster.create(Service(
            metadata=ObjectMeta(name="pool-epp", namespace="ns"),
            selector={"app": "epp"},
            ports=[
                ServicePort(name="grpc", port=9002),
                ServicePort(name="epp-metrics", port=9091),  # "metric" hit
            ],
        ))
        cluster.create(InferencePool(
            metadata=ObjectMeta(name="pool", namespace="ns"),
            selector={"app": "vllm"},
            epp_service_name="pool-epp",
        ))
        rec.reconcile("ns", "pool")
        pool = ds.pool_get("ns", "pool")
        assert pool is not None
        assert pool.selector == {"app": "vllm"}
        assert pool.endpoint_picker.service_name == "pool-epp"
        # port named *metric* wins over other ports (pool.go:117)
        assert pool.endpoint_picker.metrics_port_number == 9091

    def test_missing_service_falls_back_to_9090(self):
        from wva_amd.kube.objects import InferencePool

        cluster, ds, rec = self._stack()
        cluster.create(InferencePool(
            metadata=ObjectMeta(name="pool", namespace="ns"),
            selector={"app": "vllm"},
            epp_service_name="absent-svc",
        ))
        rec.reconcile("ns", "pool")
        assert ds.pool_get("ns", "pool").endpoint_picker \
            .metrics_port_number == 9090

    def test_default_epp_service_name_convention(self):
        from wva_amd.kube.objects import InferencePool

        cluster, ds, rec = self._stack()
        cluster.create(InferencePool(
            metadata=ObjectMeta(name="mypool", namespace="ns"),
            selector={"a": "b"},
        ))
        rec.reconcile("ns", "mypool")
        assert ds.pool_get("ns", "mypool").endpoint_picker \
            .service_name == "mypool-epp"

    def test_deletion_removes_pool_and_source(self):
        from wva_amd.collector.registry import SourceRegistry
        from wva_amd.controllers.inferencepool import InferencePoolReconciler
        from wva_amd.datastore.datastore import Datastore
        from wva_amd.kube.objects import InferencePool

        cluster = FakeCluster()
        registry = SourceRegistry()
        ds = Datastore(cluster, source_registry=registry)
        rec = InferencePoolReconciler(cluster, ds)
        cluster.create(InferencePool(
            metadata=ObjectMeta(name="pool", namespace="ns"),
            selector={"a": "b"},
        ))
        rec.reconcile("ns", "pool")
        assert ds.pool_get("ns", "pool") is not None
        src = ds.pool_source("ns", "pool")
        assert src is not None and registry.get(src.name()) is not None

        cluster.delete("InferencePool", "ns", "pool")
        rec.reconcile("ns", "pool")
        assert ds.pool_get("ns", "pool") is None
        assert registry.get(src.name()) is None


class TestDedupWorkQueue:
    def test_burst_collapses_to_one(self):
        from wva_amd.runtime.manager import DedupWorkQueue

        q = DedupWorkQueue()
        fn = lambda ns, n: None  # noqa: E731
        for _ in range(50):
            q.put((fn, "ns", "a"))
        q.put((fn, "ns", "b"))
        assert q.get(0.1) == (fn, "ns", "a")
        assert q.get(0.1) == (fn, "ns", "b")
        assert q.get(0.05) is None  # the 49 duplicates collapsed

    def test_readd_after_pickup_allowed(self):
        from wva_amd.runtime.manager import DedupWorkQueue

        q = DedupWorkQueue()
        fn = lambda ns, n: None  # noqa: E731
        q.put((fn, "ns", "a"))
        item = q.get(0.1)
        q.put(item)  # re-add while "processing" — must queue again
        assert q.get(0.1) == item

    def test_fifo_order_preserved(self):
        from wva_amd.runtime.manager import DedupWorkQueue

        q = DedupWorkQueue()
        fn = lambda ns, n: None  # noqa: E731
        for name in ("x", "y", "z"):
            q.put((fn, "ns", name))
        assert [q.get(0.1)[2] for _ in range(3)] == ["x", "y", "z"]


class TestLeaderReleaseRobustness:
    def test_release_swallows_conflict(self):
        """ReleaseOnCancel is best-effort: a competitor's concurrent
        lease write (409) must not abort Manager.stop()."""
        from wva_amd.kube.fake import ConflictError, FakeCluster
        from wva_amd.runtime.manager import LeaderElector

        cluster = FakeCluster()
        el = LeaderElector(cluster, "wva-lease", identity="me")
        assert el.try_acquire_or_renew()

        class Conflicting:
            def try_get(self, *a):
                return cluster.try_get(*a)

            def update(self, obj, **kw):
                raise ConflictError("409")

        el.cluster = Conflicting()
        el.release()  # must not raise

    def test_release_swallows_connection_error(self):
        from wva_amd.runtime.manager import LeaderElector

        class Down:
            def try_get(self, *a):
                raise OSError("connection refused")

        el = LeaderElector(Down(), "wva-lease", identity="me")
        el.release()  # must not raise


class TestReconcilerConflictRetry:
    def test_status_conflict_retried_against_fresh_read(self):
        """A competing writer bumping the VA between the reconciler's
        read and its status write (409) must not drop the decision —
        one retry against a fresh read persists it."""
        from wva_amd.analyzers.interfaces import VariantDecision
        from wva_amd.api.types import (
            CrossVersionObjectReference, ObjectMeta, VariantAutoscaling,
            VariantAutoscalingSpec, utcnow,
        )
        from wva_amd.controllers.variantautoscaling import (
            VariantAutoscalingReconciler,
        )
        from wva_amd.datastore.datastore import Datastore
        from wva_amd.engines.common import DecisionCache
        from wva_amd.kube.fake import FakeCluster
        from wva_amd.kube.objects import Deployment

        cluster = FakeCluster()
        cluster.create(Deployment(
            metadata=ObjectMeta(name="d", namespace="ns"),
        ))
        cluster.create(VariantAutoscaling(
            metadata=ObjectMeta(name="va", namespace="ns"),
            spec=VariantAutoscalingSpec(
                scale_target_ref=CrossVersionObjectReference(name="d"),
                model_id="m",
            ),
        ))
        cache = DecisionCache()
        cache.set("ns", "va", VariantDecision(
            variant_name="va", namespace="ns",
            accelerator_name="MI355X", target_replicas=3,
            last_run_time=utcnow(), metrics_available=True,
        ))
        rec = VariantAutoscalingReconciler(
            cluster, Datastore(cluster), cache,
        )

        # simulate a precondition-enforcing status write: the first
        # attempt 409s (competitor won the race and also relabeled the
        # VA); the retry against a fresh read must persist the decision
        from wva_amd.kube.fake import ConflictError

        real_update_status = cluster.update_status
        calls = {"n": 0}

        def conflicting(obj):
            if calls["n"] == 0:
                calls["n"] += 1
                competitor = cluster.get("VariantAutoscaling", "ns", "va")
                competitor.metadata.labels["competitor"] = "1"
                cluster.update(competitor)
                raise ConflictError("simulated 409")
            return real_update_status(obj)

        cluster.update_status = conflicting
        rec.reconcile("ns", "va")
        assert calls["n"] == 1  # the conflict branch actually fired
        va = cluster.get("VariantAutoscaling", "ns", "va")
        assert va.status.desired_optimized_alloc.num_replicas == 3
        assert va.metadata.labels.get("competitor") == "1"  # both writes held
