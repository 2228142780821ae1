"""Threaded stress tests for every mutex-guarded shared structure —
the Python analog of SURVEY §5's "add -race to unit CI" recommendation.

Each test hammers one structure from many threads and asserts (a) no
exception escapes a worker, (b) the structure's invariants hold after
the storm. CPython's GIL doesn't serialize compound operations, so
missing locks DO corrupt these structures (lost updates, dict-resize
RuntimeError during iteration) — these tests catch a dropped lock.
"""
import threading

import pytest

N_THREADS = 8
N_OPS = 300


def _storm(worker, n_threads=N_THREADS):
    """Run worker(tid) across threads; re-raise the first exception."""
    errors = []

    def run(tid):
        try:
            worker(tid)
        except Exception as e:  # noqa: BLE001 — surfaced below
            errors.append(e)

    threads = [
        threading.Thread(target=run, args=(t,)) for t in range(n_threads)
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=30)
        assert not t.is_alive(), "stress worker deadlocked"
    if errors:
        raise errors[0]


class TestDecisionCacheUnderStress:
    def test_concurrent_set_get_delete(self):
        from wva_amd.analyzers.interfaces import VariantDecision
        from wva_amd.engines.common import DecisionCache

        cache = DecisionCache()

        def worker(tid):
            for i in range(N_OPS):
                name = f"va-{i % 10}"
                cache.set("ns", name, VariantDecision(
                    variant_name=name, target_replicas=tid,
                ))
                got = cache.get("ns", name)
                assert got is None or isinstance(got, VariantDecision)
                if i % 7 == 0:
                    cache.delete("ns", name)
                len(cache)

        _storm(worker)
        # survivors are intact decisions
        for i in range(10):
            d = cache.get("ns", f"va-{i}")
            assert d is None or 0 <= d.target_replicas < N_THREADS


class TestCapacityStoreUnderStress:
    def test_concurrent_update_find_evict(self):
        from wva_amd.analyzers.capacity_store import (
            CapacityKnowledgeStore, CapacityRecord,
        )
        from wva_amd.analyzers.deployment_parser import VLLMEngineParams

        store = CapacityKnowledgeStore()

        def worker(tid):
            for i in range(N_OPS):
                v = f"v{i % 5}"
                store.update("ns", "m", v, CapacityRecord(
                    accelerator_name="MI355X", gpu_count=1,
                    total_kv_capacity_tokens=1000 + tid,
                    effective_capacity=900 + tid,
                    vllm_params=VLLMEngineParams(),
                    learned_from="live",
                ))
                store.get("ns", "m", v)
                store.find_compatible("m", "MI355X", 1, VLLMEngineParams())
                store.is_stale("ns", "m", v)
                if i % 50 == 0:
                    store.evict_stale(timeout_seconds=3600.0)
                len(store)

        _storm(worker)
        assert len(store) == 5
        for i in range(5):
            rec = store.get("ns", "m", f"v{i}")
            assert rec is not None
            assert rec.total_kv_capacity_tokens - 1000 == \
                rec.effective_capacity - 900  # one thread's write, intact


class TestConfigUnderStress:
    def test_concurrent_config_reload_and_read(self):
        from wva_amd.config.config import Config
        from wva_amd.config.saturation import SaturationScalingConfig

        cfg = Config()

        def worker(tid):
            ns = f"ns-{tid % 3}"
            for i in range(N_OPS):
                c = SaturationScalingConfig(analyzer_name="saturation")
                c.apply_defaults()
                c.kv_cache_threshold = 0.5 + (tid % 5) / 10.0
                if i % 2 == 0:
                    cfg.update_saturation_config(c)
                else:
                    cfg.update_saturation_config_for_namespace(ns, c)
                out = cfg.saturation_config_for_namespace(ns)
                assert 0.5 <= out.kv_cache_threshold <= 0.9
                cfg.set_scale_to_zero_enabled(tid % 2 == 0)
                cfg.scale_to_zero_enabled()
                if i % 41 == 0:
                    cfg.remove_saturation_config_for_namespace(ns)

        _storm(worker)
        assert cfg.saturation_config() is not None


class TestQueryListUnderStress:
    def test_concurrent_register_render(self):
        from wva_amd.collector.query_template import (
            QueryList, QueryTemplate,
        )

        ql = QueryList()

        def worker(tid):
            for i in range(N_OPS):
                name = f"q-{tid}-{i % 5}"
                try:
                    ql.register(QueryTemplate(
                        name=name,
                        template='m{ns="{{.ns}}"}',
                        params=["ns"],
                    ))
                except ValueError:
                    pass  # duplicate registration raises by contract
                t = ql.get(name)
                assert t.render({"ns": f"n{tid}"}) == f'm{{ns="n{tid}"}}'
                ql.names()
                ql.has(name)

        _storm(worker)


class TestTTLCacheUnderStress:
    def test_concurrent_put_get_expire(self):
        from wva_amd.collector.cache import TTLCache

        cache = TTLCache(ttl_seconds=0.01)

        def worker(tid):
            for i in range(N_OPS):
                key = f"k{i % 20}"
                cache.put(key, [tid, i])
                got = cache.get(key)
                assert got is None or isinstance(got, list)

        _storm(worker)


class TestDatastoreUnderStress:
    def test_concurrent_pool_registration(self):
        from wva_amd.datastore.datastore import Datastore
        from wva_amd.kube.fake import FakeCluster
        from wva_amd.kube.objects import EndpointPool, EndpointPicker

        ds = Datastore(FakeCluster())

        def worker(tid):
            for i in range(N_OPS // 3):
                pool = EndpointPool(
                    name=f"pool-{i % 5}", namespace="ns",
                    selector={"app": f"a{tid}"},
                    endpoint_picker=EndpointPicker(
                        service_name=f"svc{tid}", namespace="ns",
                    ),
                )
                ds.pool_set(pool)
                ds.pool_get("ns", f"pool-{i % 5}")
                ds.namespace_track(f"ns-{tid}")
                ds.tracked_namespaces()
                ds.pools()
                if i % 29 == 0:
                    ds.pool_delete("ns", f"pool-{i % 5}")

        _storm(worker)
        assert 4 <= len(ds.pools()) <= 5


class TestApiServerUnderConcurrentClients:
    """The in-process API server (tests/k8s_test_server.py) backs every
    REST/e2e suite; hammer it with parallel writers + a watcher and
    check nothing is lost, duplicated, or 500'd."""

    def test_parallel_writers_converge(self):
        import sys
        sys.path.insert(0, "tests")
        from k8s_test_server import K8sTestServer
        from wva_amd.api.types import ObjectMeta
        from wva_amd.kube.fake import ConflictError, FakeCluster
        from wva_amd.kube.objects import ConfigMap
        from wva_amd.kube.rest import RestCluster

        backing = FakeCluster()
        server = K8sTestServer(backing).start()
        clients = [RestCluster(server.url) for _ in range(4)]
        try:
            def worker(tid):
                c = clients[tid]
                for i in range(40):
                    name = f"cm-{tid}-{i % 5}"
                    cm = c.try_get("ConfigMap", "ns", name)
                    if cm is None:
                        try:
                            c.create(ConfigMap(
                                metadata=ObjectMeta(name=name, namespace="ns"),
                                data={"v": "0"},
                            ))
                        except Exception:
                            pass  # lost the create race
                    else:
                        cm.data["v"] = str(i)
                        try:
                            c.update(cm)
                        except ConflictError:
                            pass  # optimistic-concurrency loss; next loop
            _storm(worker, n_threads=4)

            # convergence: exactly the distinct names exist, readable by
            # every client, each fully-formed
            names = {f"cm-{t}-{i}" for t in range(4) for i in range(5)}
            listed = {o.metadata.name
                      for o in clients[0].list("ConfigMap", namespace="ns")}
            assert listed == names
            for c in clients:
                obj = c.get("ConfigMap", "ns", "cm-0-0")
                assert "v" in obj.data
        finally:
            for c in clients:
                c.close()
            server.stop()

    def test_watcher_sees_every_create_during_write_storm(self):
        import sys, time as _time
        sys.path.insert(0, "tests")
        from k8s_test_server import K8sTestServer
        from wva_amd.api.types import ObjectMeta
        from wva_amd.kube.fake import FakeCluster
        from wva_amd.kube.objects import ConfigMap
        from wva_amd.kube.rest import RestCluster

        backing = FakeCluster()
        server = K8sTestServer(backing).start()
        watcher = RestCluster(server.url)
        writer = RestCluster(server.url)
        try:
            q = watcher.watch(["ConfigMap"])

            def worker(tid):
                for i in range(25):
                    writer.create(ConfigMap(
                        metadata=ObjectMeta(
                            name=f"w{tid}-{i}", namespace="ns",
                        ),
                        data={},
                    ))
            _storm(worker, n_threads=3)

            expected = {f"w{t}-{i}" for t in range(3) for i in range(25)}
            seen = set()
            deadline = _time.time() + 20
            while _time.time() < deadline and not expected <= seen:
                try:
                    evt = q.get(timeout=0.25)
                except Exception:
                    continue
                if evt.type == "ADDED" and evt.obj is not None:
                    seen.add(evt.obj.metadata.name)
            assert expected <= seen  # no create lost on the stream
            watcher.stop_watch(q)
        finally:
            watcher.close()
            writer.close()
            server.stop()
