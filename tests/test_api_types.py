"""Tests for the VariantAutoscaling API types and condition helpers.

Reference parity target: api/v1alpha1/variantautoscaling_types.go and
conditions.go behavior.
"""
import time

from wva_amd.api import conditions
from wva_amd.api.types import (
    TYPE_METRICS_AVAILABLE,
    TYPE_TARGET_RESOLVED,
    CrossVersionObjectReference,
    ObjectMeta,
    OptimizedAlloc,
    VariantAutoscaling,
    VariantAutoscalingSpec,
    utcnow,
)


def make_va(name="vllm-llama", ns="default", model="meta-llama/Llama-3.1-8B"):
    return VariantAutoscaling(
        metadata=ObjectMeta(name=name, namespace=ns),
        spec=VariantAutoscalingSpec(
            scale_target_ref=CrossVersionObjectReference(name=name),
            model_id=model,
        ),
    )


class TestSpec:
    def test_defaults(self):
        va = make_va()
        assert va.spec.variant_cost == "10.0"
        assert va.spec.cost() == 10.0
        assert va.get_scale_target_kind() == "Deployment"
        assert va.get_scale_target_api() == "apps/v1"
        assert va.get_scale_target_name() == "vllm-llama"

    def test_validation(self):
        va = make_va()
        assert va.spec.validate() == []
        va.spec.model_id = ""
        assert any("modelID" in e for e in va.spec.validate())
        va.spec.variant_cost = "abc"
        assert any("variantCost" in e for e in va.spec.validate())
        assert va.spec.cost() == 10.0  # fallback on bad input

    def test_variant_cost_pattern(self):
        va = make_va()
        for good in ("10", "10.0", "0.5", "100.25"):
            va.spec.variant_cost = good
            assert not [e for e in va.spec.validate() if "variantCost" in e]
        for bad in ("-1", "1e3", "ten", "1.", ".5"):
            va.spec.variant_cost = bad
            assert [e for e in va.spec.validate() if "variantCost" in e]

    def test_roundtrip(self):
        va = make_va()
        va.status.desired_optimized_alloc = OptimizedAlloc(
            last_run_time=utcnow(), accelerator="MI355X", num_replicas=3
        )
        conditions.set_condition(
            va, TYPE_TARGET_RESOLVED, "True", "TargetFound", "ok"
        )
        d = va.to_dict()
        assert d["apiVersion"] == "llmd.ai/v1alpha1"
        assert d["kind"] == "VariantAutoscaling"
        assert d["spec"]["modelID"] == "meta-llama/Llama-3.1-8B"
        assert d["status"]["desiredOptimizedAlloc"]["numReplicas"] == 3
        va2 = VariantAutoscaling.from_dict(d)
        assert va2.spec.model_id == va.spec.model_id
        assert va2.status.desired_optimized_alloc.accelerator == "MI355X"
        assert va2.status.conditions[0].type == TYPE_TARGET_RESOLVED
        assert va2.full_name() == "vllm-llama:default"


class TestConditions:
    def test_set_and_get(self):
        va = make_va()
        conditions.set_condition(va, TYPE_METRICS_AVAILABLE, "True", "MetricsFound")
        c = conditions.get_condition(va, TYPE_METRICS_AVAILABLE)
        assert c is not None and c.status == "True"
        assert conditions.is_condition_true(va, TYPE_METRICS_AVAILABLE)
        assert not conditions.is_condition_false(va, TYPE_METRICS_AVAILABLE)

    def test_transition_time_only_on_status_change(self):
        va = make_va()
        conditions.set_condition(va, TYPE_METRICS_AVAILABLE, "True", "MetricsFound")
        t1 = conditions.get_condition(va, TYPE_METRICS_AVAILABLE).last_transition_time
        time.sleep(0.01)
        conditions.set_condition(va, TYPE_METRICS_AVAILABLE, "True", "MetricsFound2")
        t2 = conditions.get_condition(va, TYPE_METRICS_AVAILABLE).last_transition_time
        assert t1 == t2  # same status → no transition-time bump
        conditions.set_condition(va, TYPE_METRICS_AVAILABLE, "False", "MetricsMissing")
        t3 = conditions.get_condition(va, TYPE_METRICS_AVAILABLE).last_transition_time
        assert t3 >= t1

    def test_observed_generation_tracks(self):
        va = make_va()
        va.metadata.generation = 7
        conditions.set_condition(va, TYPE_METRICS_AVAILABLE, "True", "MetricsFound")
        assert (
            conditions.get_condition(va, TYPE_METRICS_AVAILABLE).observed_generation
            == 7
        )
