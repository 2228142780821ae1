"""RestCluster (kube/rest.py) against the in-process minimal API server
(tests/k8s_test_server.py) — the REST path's envtest analog. Exercises
real HTTP round trips through the serde layer for every verb the
controller uses, plus the chunked watch stream.
"""
import queue
import time

import pytest

from wva_amd.api.types import (
    CrossVersionObjectReference,
    ObjectMeta,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.kube.fake import ADDED, DELETED, MODIFIED, NotFoundError
from wva_amd.kube.objects import (
    ConfigMap,
    Container,
    Deployment,
    Node,
    Pod,
    PodStatus,
    PodTemplateSpec,
)
from wva_amd.kube.rest import RestCluster

from k8s_test_server import K8sTestServer


@pytest.fixture()
def server():
    srv = K8sTestServer().start()
    yield srv
    srv.stop()


@pytest.fixture()
def client(server):
    c = RestCluster(server.url)
    yield c
    c.close()


def make_deployment(name="vllm-d", ns="default", replicas=2):
    return Deployment(
        metadata=ObjectMeta(name=name, namespace=ns, labels={"app": name}),
        replicas=replicas,
        selector={"app": name},
        template=PodTemplateSpec(
            labels={"app": name},
            containers=[Container(
                args=["--max-num-seqs", "256"],
                requests={"amd.com/gpu": "1"},
            )],
        ),
    )


class TestCrud:
    def test_create_get_roundtrip(self, client):
        client.create(make_deployment())
        d = client.get("Deployment", "default", "vllm-d")
        assert d.replicas == 2
        assert d.template.containers[0].args == ["--max-num-seqs", "256"]
        assert d.template.containers[0].requests == {"amd.com/gpu": "1"}

    def test_get_missing_raises(self, client):
        with pytest.raises(NotFoundError):
            client.get("Deployment", "default", "nope")
        assert client.try_get("Deployment", "default", "nope") is None

    def test_list_with_label_selector(self, client):
        client.create(make_deployment("a"))
        client.create(make_deployment("b"))
        all_ = client.list("Deployment", namespace="default")
        assert {d.name for d in all_} == {"a", "b"}
        only = client.list(
            "Deployment", namespace="default", label_selector={"app": "a"}
        )
        assert [d.name for d in only] == ["a"]

    def test_update(self, client):
        client.create(make_deployment())
        d = client.get("Deployment", "default", "vllm-d")
        d.replicas = 5
        client.update(d)
        assert client.get("Deployment", "default", "vllm-d").replicas == 5

    def test_delete(self, client):
        client.create(make_deployment())
        client.delete("Deployment", "default", "vllm-d")
        assert client.try_get("Deployment", "default", "vllm-d") is None
        with pytest.raises(NotFoundError):
            client.delete("Deployment", "default", "vllm-d")

    def test_scale_subresource(self, client):
        client.create(make_deployment(replicas=1))
        d = client.scale("Deployment", "default", "vllm-d", 4)
        assert d.replicas == 4

    def test_node_cluster_scoped(self, client):
        client.create(Node(
            metadata=ObjectMeta(
                name="mi355x-0",
                labels={"amd.com/gpu.product": "AMD-Instinct-MI355X-288GB"},
            ),
            allocatable={"amd.com/gpu": "8"},
        ))
        nodes = client.list("Node")
        assert nodes[0].allocatable == {"amd.com/gpu": "8"}
        assert nodes[0].labels["amd.com/gpu.product"].endswith("288GB")

    def test_configmap(self, client):
        client.create(ConfigMap(
            metadata=ObjectMeta(
                name="wva-saturation-scaling-config", namespace="wva-system"
            ),
            data={"kvCacheThreshold": "0.8"},
        ))
        cm = client.get(
            "ConfigMap", "wva-system", "wva-saturation-scaling-config"
        )
        assert cm.data["kvCacheThreshold"] == "0.8"

    def test_pod_status_roundtrip(self, client):
        client.create(Pod(
            metadata=ObjectMeta(name="p1", namespace="default",
                                labels={"app": "x"}),
            containers=[Container()],
            status=PodStatus(phase="Running", ready=True, pod_ip="10.0.0.9"),
        ))
        p = client.get("Pod", "default", "p1")
        assert p.is_ready()
        assert p.status.pod_ip == "10.0.0.9"


class TestVariantAutoscaling:
    def va(self):
        return VariantAutoscaling(
            metadata=ObjectMeta(
                name="vllm-d", namespace="default",
                labels={"inference.optimization/acceleratorName": "MI355X"},
            ),
            spec=VariantAutoscalingSpec(
                scale_target_ref=CrossVersionObjectReference(name="vllm-d"),
                model_id="meta-llama/Llama-3.1-8B",
            ),
        )

    def test_va_roundtrip(self, client):
        client.create(self.va())
        va = client.get("VariantAutoscaling", "default", "vllm-d")
        assert va.spec.model_id == "meta-llama/Llama-3.1-8B"
        assert va.metadata.labels[
            "inference.optimization/acceleratorName"] == "MI355X"

    def test_va_status_subresource(self, client):
        client.create(self.va())
        va = client.get("VariantAutoscaling", "default", "vllm-d")
        va.status.desired_optimized_alloc.num_replicas = 3
        va.status.desired_optimized_alloc.accelerator = "MI355X"
        client.update_status(va)
        got = client.get("VariantAutoscaling", "default", "vllm-d")
        assert got.status.desired_optimized_alloc.num_replicas == 3


class TestWatch:
    def _drain_until(self, q, pred, timeout=15.0):
        deadline = time.time() + timeout
        seen = []
        while time.time() < deadline:
            try:
                evt = q.get(timeout=0.2)
            except queue.Empty:
                continue
            seen.append(evt)
            if pred(evt):
                return evt, seen
        raise AssertionError(f"no matching event; saw {seen}")

    def test_watch_stream(self, client):
        client.create(make_deployment("pre"))
        q = client.watch(["Deployment"])
        # synthetic ADDED for pre-existing object (informer list phase)
        evt, _ = self._drain_until(
            q, lambda e: e.type == ADDED and e.obj.name == "pre"
        )
        assert evt.obj.replicas == 2

        client.create(make_deployment("post"))
        self._drain_until(
            q, lambda e: e.type == ADDED and e.obj.name == "post"
        )

        d = client.get("Deployment", "default", "post")
        d.replicas = 7
        client.update(d)
        evt, _ = self._drain_until(
            q, lambda e: e.type == MODIFIED and e.obj.name == "post"
            and e.obj.replicas == 7
        )

        client.delete("Deployment", "default", "post")
        self._drain_until(
            q, lambda e: e.type == DELETED and e.obj.name == "post"
        )
        client.stop_watch(q)


class TestFullStackOverRest:
    """The whole controller (build_app) operating through RestCluster:
    the emulated cluster state lives in the API server's backing store;
    the controller reads/writes it ONLY via HTTP. Proves every component
    (engine, reconcilers, collector mapping, actuator, discovery) sticks
    to the cluster client surface."""

    def test_engine_tick_over_rest(self, server):
        from prometheus_client import CollectorRegistry
        from wva_amd.app import build_app
        from wva_amd.config.config import Config
        from wva_amd.config.saturation import SaturationScalingConfig
        from wva_amd.emulator.cluster_sim import ClusterSim
        from wva_amd.emulator.sim_source import SimMetricsSource
        
        rest = RestCluster(server.url)
        try:
            # cluster "reality" lives in the backing FakeCluster; the sim
            # manipulates it directly (pods, metrics), the controller sees
            # it only through REST
            backing = server.cluster
            rest.create(Node(
                metadata=ObjectMeta(
                    name="mi355x-0",
                    labels={
                        "amd.com/gpu.product": "AMD-Instinct-MI355X-288GB",
                        "amd.com/gpu.memory": "294912",
                    },
                ),
                allocatable={"amd.com/gpu": "8"},
            ))
            rest.create(make_deployment("vllm-d", replicas=1))
            rest.create(VariantAutoscaling(
                metadata=ObjectMeta(
                    name="vllm-d", namespace="default",
                    labels={
                        "inference.optimization/acceleratorName": "MI355X"
                    },
                ),
                spec=VariantAutoscalingSpec(
                    scale_target_ref=CrossVersionObjectReference(
                        name="vllm-d"
                    ),
                    model_id="meta-llama/Llama-3.1-8B",
                ),
            ))

            from wva_amd.emulator.vllm_sim import ServiceProfile
            sim = ClusterSim(backing, warm_start=True)
            sim.register_variant(
                "meta-llama/Llama-3.1-8B", "default", "vllm-d",
                ServiceProfile(),
            )
            sim.reconcile_deployments()
            # drive some load so metrics are non-trivial
            model = sim.model("meta-llama/Llama-3.1-8B", "default")
            for _ in range(40):
                sim.generate_arrivals(
                    model, lambda t: 50.0, 0.25, 100, 50
                )
                sim.advance(0.25)

            config = Config()
            config.update_saturation_config(
                SaturationScalingConfig.from_dict(
                    {"analyzerName": "saturation"}
                )
            )
            config.mark_bootstrap_complete()
            app = build_app(
                rest, config, source=SimMetricsSource(sim),
                metrics_registry=CollectorRegistry(), start_engines=False,
            )
            app.saturation_engine.optimize()
            app.va_reconciler.reconcile("default", "vllm-d")

            va = rest.get("VariantAutoscaling", "default", "vllm-d")
            assert va.status.desired_optimized_alloc.num_replicas >= 1
            assert va.status.desired_optimized_alloc.accelerator == "MI355X"
            conds = {c.type for c in va.status.conditions}
            assert "OptimizationReady" in conds
        finally:
            rest.close()


class TestLeaderElectionOverRest:
    def test_lease_acquire_renew_release(self, server):
        from wva_amd.runtime.manager import LeaderElector

        a = RestCluster(server.url)
        b = RestCluster(server.url)
        try:
            e1 = LeaderElector(a, "wva-lock", identity="pod-a",
                               lease_duration=60)
            e2 = LeaderElector(b, "wva-lock", identity="pod-b",
                               lease_duration=60)
            assert e1.try_acquire_or_renew() is True
            assert e2.try_acquire_or_renew() is False  # held by pod-a
            assert e1.try_acquire_or_renew() is True   # renew
            e1.release()
            assert e2.try_acquire_or_renew() is True   # fast failover
        finally:
            a.close()
            b.close()


class TestWatchSince:
    def test_no_event_lost_between_list_and_watch(self, server):
        """The LIST→WATCH gap: mutations between a snapshot and the watch
        subscription must be replayed via resourceVersion (FakeCluster
        watch_since; the API-server watch-cache contract). Without it,
        events racing the stream establishment were silently lost."""
        from wva_amd.kube.fake import ADDED

        backing = server.cluster
        backing.create(make_deployment("a"))
        objs, rv = backing.snapshot("Deployment")
        assert len(objs) == 1
        # mutation AFTER snapshot, BEFORE subscription
        backing.create(make_deployment("raced"))
        q = backing.watch_since(["Deployment"], rv)
        evt = q.get(timeout=2)
        assert evt.type == ADDED and evt.obj.name == "raced"
        # live events still flow
        backing.create(make_deployment("live"))
        evt = q.get(timeout=2)
        assert evt.obj.name == "live"
        backing.stop_watch(q)
