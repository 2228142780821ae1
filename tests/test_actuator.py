"""Actuator + DirectActuator suite (reference internal/actuator tests,
830 LoC envtest — the replica-count fallback chain, gauge emission
semantics, failure tolerance, and the scale-subresource writer, here
run against both FakeCluster and the REST path).
"""
import pytest
from prometheus_client import CollectorRegistry

from wva_amd.actuator.actuator import Actuator
from wva_amd.actuator.direct import DirectActuator
from wva_amd.api.types import (
    CrossVersionObjectReference,
    ObjectMeta,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.kube.fake import FakeCluster, NotFoundError
from wva_amd.kube.objects import Deployment, DeploymentStatus, PodTemplateSpec
from wva_amd.metrics.metrics import MetricsEmitter

NS = "default"


def make_va(name="v", replicas_desired=3, accel="MI355X"):
    va = VariantAutoscaling(
        metadata=ObjectMeta(name=name, namespace=NS),
        spec=VariantAutoscalingSpec(
            scale_target_ref=CrossVersionObjectReference(name=name),
            model_id="m",
        ),
    )
    va.status.desired_optimized_alloc.num_replicas = replicas_desired
    va.status.desired_optimized_alloc.accelerator = accel
    return va


def make_deploy(cluster, name="v", spec_replicas=2, status_replicas=0):
    d = Deployment(
        metadata=ObjectMeta(name=name, namespace=NS),
        replicas=spec_replicas,
        template=PodTemplateSpec(labels={"app": name}),
        status=DeploymentStatus(replicas=status_replicas),
    )
    cluster.create(d)
    return d


def gauge_value(registry, metric, **labels):
    for family in registry.collect():
        if family.name != metric:
            continue
        for sample in family.samples:
            if all(sample.labels.get(k) == v for k, v in labels.items()):
                return sample.value
    return None


class TestReplicaCountFallbackChain:
    """actuator.go:21-29: status → spec → 1 (engine.go:525-528 semantics:
    zero status with non-zero spec means the controller lags)."""

    def test_status_wins(self):
        c = FakeCluster()
        make_deploy(c, spec_replicas=2, status_replicas=5)
        a = Actuator(c, MetricsEmitter(registry=CollectorRegistry()))
        assert a.get_current_deployment_replicas(make_va()) == 5

    def test_spec_fallback_when_status_zero(self):
        c = FakeCluster()
        make_deploy(c, spec_replicas=2, status_replicas=0)
        a = Actuator(c, MetricsEmitter(registry=CollectorRegistry()))
        assert a.get_current_deployment_replicas(make_va()) == 2

    def test_genuinely_zero(self):
        c = FakeCluster()
        make_deploy(c, spec_replicas=0, status_replicas=0)
        a = Actuator(c, MetricsEmitter(registry=CollectorRegistry()))
        assert a.get_current_deployment_replicas(make_va()) == 0

    def test_missing_deployment_raises_after_retry(self):
        c = FakeCluster()
        a = Actuator(c, MetricsEmitter(registry=CollectorRegistry()))
        with pytest.raises(NotFoundError):
            a.get_current_deployment_replicas(make_va())


class TestEmitMetrics:
    def test_gauges_emitted_with_labels(self):
        c = FakeCluster()
        make_deploy(c, status_replicas=2)
        reg = CollectorRegistry()
        a = Actuator(c, MetricsEmitter(registry=reg))
        a.emit_metrics(make_va(replicas_desired=4))
        labels = dict(variant_name="v", namespace=NS,
                      accelerator_type="MI355X")
        assert gauge_value(reg, "wva_current_replicas", **labels) == 2
        assert gauge_value(reg, "wva_desired_replicas", **labels) == 4
        assert gauge_value(reg, "wva_desired_ratio", **labels) == 2.0

    def test_ratio_zero_current_is_desired(self):
        """metrics.go:114-165: ratio with current=0 reports desired
        (avoid div-by-zero starving HPA)."""
        c = FakeCluster()
        make_deploy(c, spec_replicas=0, status_replicas=0)
        reg = CollectorRegistry()
        a = Actuator(c, MetricsEmitter(registry=reg))
        a.emit_metrics(make_va(replicas_desired=3))
        assert gauge_value(
            reg, "wva_desired_ratio",
            variant_name="v", namespace=NS, accelerator_type="MI355X",
        ) == 3.0

    def test_negative_desired_not_emitted(self):
        c = FakeCluster()
        make_deploy(c)
        reg = CollectorRegistry()
        a = Actuator(c, MetricsEmitter(registry=reg))
        a.emit_metrics(make_va(replicas_desired=-1))
        assert gauge_value(
            reg, "wva_desired_replicas",
            variant_name="v", namespace=NS, accelerator_type="MI355X",
        ) is None

    def test_missing_deployment_emits_current_zero(self):
        """Emission failure tolerance (actuator.go: never fail the
        tick): a missing target emits current=0, keeping the signal."""
        c = FakeCluster()
        reg = CollectorRegistry()
        a = Actuator(c, MetricsEmitter(registry=reg))
        a.emit_metrics(make_va(replicas_desired=2))
        assert gauge_value(
            reg, "wva_current_replicas",
            variant_name="v", namespace=NS, accelerator_type="MI355X",
        ) == 0


class TestDirectActuator:
    def test_scale_writes_spec_replicas(self):
        c = FakeCluster()
        make_deploy(c, spec_replicas=0)
        DirectActuator(c).scale_target_object("Deployment", NS, "v", 1)
        assert c.get("Deployment", NS, "v").replicas == 1

    def test_scale_missing_target_raises(self):
        c = FakeCluster()
        with pytest.raises(NotFoundError):
            DirectActuator(c).scale_target_object("Deployment", NS, "x", 1)

    def test_scale_over_rest(self):
        """The 0→1 transition through the real scale subresource
        endpoint (direct_actuator.go:78-104 over HTTP)."""
        import sys
        sys.path.insert(0, "tests")
        from k8s_test_server import K8sTestServer
        from wva_amd.kube.rest import RestCluster

        backing = FakeCluster()
        make_deploy(backing, spec_replicas=0)
        server = K8sTestServer(backing).start()
        try:
            rest = RestCluster(server.url)
            DirectActuator(rest).scale_target_object("Deployment", NS, "v", 1)
            assert backing.get("Deployment", NS, "v").replicas == 1
            rest.close()
        finally:
            server.stop()
