"""Engine edge behavior: safety-net metric emission on analysis failure,
decision-trigger overflow, limited-mode inventory failure, variant utils.
"""
import os

from prometheus_client import CollectorRegistry

from wva_amd.api.types import (
    CrossVersionObjectReference,
    ObjectMeta,
    OptimizedAlloc,
    VariantAutoscaling,
    VariantAutoscalingSpec,
    utcnow,
)
from wva_amd.app import build_app
from wva_amd.collector.source import MetricResult, RefreshSpec
from wva_amd.config.config import Config
from wva_amd.engines.common import DecisionTrigger
from wva_amd.kube.fake import FakeCluster
from wva_amd.kube.objects import Container, Deployment, PodTemplateSpec
from wva_amd.utils.variant import (
    active_variant_autoscalings,
    group_variant_autoscaling_by_model,
    inactive_variant_autoscalings,
)


class _FailingSource:
    """KV query returns an error → collector raises → engine safety net."""

    def name(self):
        return "prometheus"

    def query_list(self):
        from wva_amd.collector.query_template import QueryList

        return QueryList()

    def refresh(self, spec: RefreshSpec):
        return {
            q: MetricResult(query=q, error=RuntimeError("prometheus down"))
            for q in spec.queries
        }

    def get(self, query, params):
        return None


def make_va_and_deploy(cluster, name="v1", ns="default", replicas=2):
    cluster.create(Deployment(
        metadata=ObjectMeta(name=name, namespace=ns),
        replicas=replicas,
        selector={"app": name},
        template=PodTemplateSpec(labels={"app": name},
                                 containers=[Container()]),
    ))
    va = VariantAutoscaling(
        metadata=ObjectMeta(
            name=name, namespace=ns,
            labels={"inference.optimization/acceleratorName": "MI355X"},
        ),
        spec=VariantAutoscalingSpec(
            scale_target_ref=CrossVersionObjectReference(name=name),
            model_id="m",
        ),
    )
    cluster.create(va)
    return va


class TestSafetyNet:
    def test_fallback_metrics_on_collect_failure(self):
        cluster = FakeCluster()
        va = make_va_and_deploy(cluster, replicas=3)
        # previous desired recorded in status
        stored = cluster.get("VariantAutoscaling", "default", "v1")
        stored.status.desired_optimized_alloc = OptimizedAlloc(
            last_run_time=utcnow(), accelerator="MI355X", num_replicas=5
        )
        cluster.update_status(stored)

        config = Config()
        config.mark_bootstrap_complete()
        app = build_app(cluster, config, source=_FailingSource(),
                        metrics_registry=CollectorRegistry(),
                        start_engines=False)
        app.saturation_engine.optimize()  # must not raise

        # safety net emitted previous desired (5) with real current (3)
        samples = {}
        for fam in app.emitter.registry.collect():
            for s in fam.samples:
                samples[s.name] = s.value
        assert samples.get("wva_desired_replicas") == 5
        assert samples.get("wva_current_replicas") == 3

    def test_fallback_uses_current_when_no_previous_desired(self):
        cluster = FakeCluster()
        make_va_and_deploy(cluster, replicas=2)
        config = Config()
        config.mark_bootstrap_complete()
        app = build_app(cluster, config, source=_FailingSource(),
                        metrics_registry=CollectorRegistry(),
                        start_engines=False)
        app.saturation_engine.optimize()
        samples = {}
        for fam in app.emitter.registry.collect():
            for s in fam.samples:
                samples[s.name] = s.value
        assert samples.get("wva_desired_replicas") == 2  # = current


class TestDecisionTrigger:
    def test_overflow_drops_not_blocks(self):
        t = DecisionTrigger(capacity=3)
        assert t.push("ns", "a")
        assert t.push("ns", "b")
        assert t.push("ns", "c")
        assert not t.push("ns", "d")  # dropped, no block
        assert len(t) == 3
        assert t.pop(timeout=0.01) == "ns/a"


class TestVariantUtils:
    def test_active_inactive_split(self):
        cluster = FakeCluster()
        make_va_and_deploy(cluster, name="up", replicas=2)
        make_va_and_deploy(cluster, name="down", replicas=0)
        active = [va.name for va in active_variant_autoscalings(cluster)]
        inactive = [va.name for va in inactive_variant_autoscalings(cluster)]
        assert active == ["up"]
        assert inactive == ["down"]

    def test_grouping_key(self):
        cluster = FakeCluster()
        make_va_and_deploy(cluster, name="a")
        make_va_and_deploy(cluster, name="b")
        groups = group_variant_autoscaling_by_model(
            active_variant_autoscalings(cluster)
        )
        assert list(groups.keys()) == ["m|default"]
        assert len(groups["m|default"]) == 2

    def test_namespace_exclusion(self):
        from wva_amd.kube.objects import ConfigMap  # any namespaced obj

        cluster = FakeCluster()
        make_va_and_deploy(cluster)

        # a Namespace object with the exclusion annotation
        class NS:
            kind = "Namespace"
            api_version = "v1"

            def __init__(self):
                self.metadata = ObjectMeta(
                    name="default", namespace="",
                    annotations={"wva.llmd.ai/exclude": "true"},
                )

        cluster.create(NS())
        assert active_variant_autoscalings(cluster) == []


class TestLimitedMode:
    def test_inventory_failure_skips_tick(self):
        cluster = FakeCluster()
        make_va_and_deploy(cluster)
        config = Config()
        config.set_limited_mode_enabled(True)
        config.mark_bootstrap_complete()
        app = build_app(cluster, config, source=_FailingSource(),
                        metrics_registry=CollectorRegistry(),
                        start_engines=False)

        class BoomInventory:
            def refresh_all(self):
                raise RuntimeError("discovery down")

        app.saturation_engine.limiter.inventory = BoomInventory()
        app.saturation_engine.optimize()  # logged + returned, no raise


class TestScaleUpLead:
    def test_lead_inflates_required_on_rising_demand(self):
        import time as _time

        from wva_amd.analyzers.interfaces import AnalyzerResult
        from wva_amd.config.saturation import SaturationScalingConfig
        from wva_amd.engines.saturation import SaturationEngine

        eng = SaturationEngine.__new__(SaturationEngine)
        eng._demand_history = {}
        cfg = SaturationScalingConfig.from_dict(
            {"analyzerName": "saturation", "scaleUpLeadSeconds": 30.0}
        )

        r1 = AnalyzerResult(total_demand=1000.0, required_capacity=0.0)
        out1 = eng._apply_scale_up_lead("m", "ns", cfg, r1)
        assert out1.required_capacity == 0.0  # no history yet

        eng._demand_history["m|ns"] = (_time.time() - 10.0, 1000.0)
        r2 = AnalyzerResult(total_demand=1500.0, required_capacity=100.0)
        out2 = eng._apply_scale_up_lead("m", "ns", cfg, r2)
        assert out2.required_capacity > 100.0
        # capped at 25% of demand / threshold
        assert out2.required_capacity <= 100.0 + 0.25 * 1500.0 / cfg.scale_up_threshold + 1e-6

    def test_lead_ignores_falling_demand_and_default_off(self):
        import time as _time

        from wva_amd.analyzers.interfaces import AnalyzerResult
        from wva_amd.config.saturation import SaturationScalingConfig
        from wva_amd.engines.saturation import SaturationEngine

        eng = SaturationEngine.__new__(SaturationEngine)
        eng._demand_history = {"m|ns": (_time.time() - 10.0, 2000.0)}
        cfg = SaturationScalingConfig.from_dict(
            {"analyzerName": "saturation", "scaleUpLeadSeconds": 30.0}
        )
        r = AnalyzerResult(total_demand=1500.0, required_capacity=50.0)
        out = eng._apply_scale_up_lead("m", "ns", cfg, r)
        assert out.required_capacity == 50.0  # falling: no inflation

        cfg0 = SaturationScalingConfig.from_dict({"analyzerName": "saturation"})
        eng._demand_history = {"m|ns": (_time.time() - 10.0, 1000.0)}
        r2 = AnalyzerResult(total_demand=1500.0, required_capacity=50.0)
        out2 = eng._apply_scale_up_lead("m", "ns", cfg0, r2)
        assert out2.required_capacity == 50.0  # default off
