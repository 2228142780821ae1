"""V1 percentage saturation analyzer tests.

Mirrors the behavioral coverage of reference
internal/saturation/analyzer_test.go (509 LoC): saturation detection,
spare-capacity averaging, scale-up triggers, scale-down simulation and
target calculation incl. transition freeze and tie-breaking.
"""
from wva_amd.analyzers.interfaces import ReplicaMetrics, VariantReplicaState
from wva_amd.analyzers.saturation_v1 import SaturationAnalyzerV1
from wva_amd.config.saturation import SaturationScalingConfig

CFG = SaturationScalingConfig()  # defaults: kv 0.80 / q 5 / spare 0.10 / 3
AN = SaturationAnalyzerV1()


def rm(pod, kv, q, variant="v1", cost=10.0, accel="MI355X"):
    return ReplicaMetrics(
        pod_name=pod,
        kv_cache_usage=kv,
        queue_length=q,
        variant_name=variant,
        accelerator_name=accel,
        cost=cost,
    )


class TestAnalyzeModelSaturation:
    def test_empty_metrics(self):
        a = AN.analyze_model_saturation("m", "ns", [], CFG)
        assert a.total_replicas == 0
        assert not a.should_scale_up
        assert not a.scale_down_safe

    def test_saturation_by_kv(self):
        a = AN.analyze_model_saturation("m", "ns", [rm("p0", 0.85, 0)], CFG)
        assert a.non_saturated_count == 0
        assert a.variant_analyses[0].saturated_replicas == ["p0"]
        # no non-saturated replicas → avg spare 0 → scale-up triggered
        assert a.should_scale_up

    def test_saturation_by_queue(self):
        a = AN.analyze_model_saturation("m", "ns", [rm("p0", 0.1, 5)], CFG)
        assert a.non_saturated_count == 0

    def test_boundary_is_saturated(self):
        # >= threshold is saturated (exact 0.80, exact queue 5)
        a = AN.analyze_model_saturation(
            "m", "ns", [rm("p0", 0.80, 0), rm("p1", 0.0, 5)], CFG
        )
        assert a.non_saturated_count == 0

    def test_spare_capacity_averaging(self):
        a = AN.analyze_model_saturation(
            "m", "ns", [rm("p0", 0.30, 1), rm("p1", 0.50, 3)], CFG
        )
        assert a.non_saturated_count == 2
        assert abs(a.avg_spare_kv_capacity - ((0.5 + 0.3) / 2)) < 1e-9
        assert abs(a.avg_spare_queue_length - ((4 + 2) / 2)) < 1e-9
        assert not a.should_scale_up

    def test_scale_up_on_low_kv_spare(self):
        # spare kv = 0.80-0.75 = 0.05 < 0.10 trigger
        a = AN.analyze_model_saturation("m", "ns", [rm("p0", 0.75, 0)], CFG)
        assert a.should_scale_up
        assert "KV spare" in a.scale_up_reason

    def test_scale_up_on_low_queue_spare(self):
        # spare queue = 5-3 = 2 < 3 trigger
        a = AN.analyze_model_saturation("m", "ns", [rm("p0", 0.10, 3)], CFG)
        assert a.should_scale_up
        assert "queue spare" in a.scale_up_reason

    def test_scale_down_requires_two_non_saturated(self):
        a = AN.analyze_model_saturation("m", "ns", [rm("p0", 0.05, 0)], CFG)
        assert not a.scale_down_safe

    def test_scale_down_safe_when_idle(self):
        a = AN.analyze_model_saturation(
            "m", "ns", [rm("p0", 0.05, 0), rm("p1", 0.05, 0), rm("p2", 0.05, 0)], CFG
        )
        assert a.scale_down_safe

    def test_scale_down_unsafe_when_loaded(self):
        # load kv=0.6 each; after removal: 0.6*2/1? with 2 non-saturated:
        # factor 2 → kv load 1.2 > threshold → unsafe
        a = AN.analyze_model_saturation(
            "m", "ns", [rm("p0", 0.60, 0), rm("p1", 0.60, 0)], CFG
        )
        assert not a.scale_down_safe

    def test_max_usage_tracked(self):
        a = AN.analyze_model_saturation(
            "m", "ns", [rm("p0", 0.2, 1), rm("p1", 0.6, 4)], CFG
        )
        v = a.variant_analyses[0]
        assert v.max_kv_cache_usage == 0.6
        assert v.max_queue_length == 4

    def test_multi_variant_grouping(self):
        a = AN.analyze_model_saturation(
            "m",
            "ns",
            [
                rm("p0", 0.2, 0, variant="a", cost=5),
                rm("p1", 0.9, 9, variant="b", cost=20),
            ],
            CFG,
        )
        assert len(a.variant_analyses) == 2
        by_name = {v.variant_name: v for v in a.variant_analyses}
        assert by_name["a"].cost == 5
        assert by_name["b"].saturated_replicas == ["p1"]


def vs(name, current, desired=0, pending=0):
    return VariantReplicaState(
        variant_name=name,
        current_replicas=current,
        desired_replicas=desired,
        pending_replicas=pending,
    )


class TestCalculateSaturationTargets:
    def _analysis(self, metrics):
        return AN.analyze_model_saturation("m", "ns", metrics, CFG)

    def test_none_analysis_defaults_to_current(self):
        targets = AN.calculate_saturation_targets(None, [vs("a", 3)])
        assert targets == {"a": 3}

    def test_stable_no_action(self):
        # kv=0.5: no scale-up (spare 0.3 > 0.1) and scale-down unsafe
        # (load 0.5 × 2 = 1.0 > 0.8 threshold after redistribution)
        a = self._analysis([rm("p0", 0.5, 0), rm("p1", 0.5, 0)])
        assert not a.should_scale_up and not a.scale_down_safe
        targets = AN.calculate_saturation_targets(a, [vs("v1", 2)])
        assert targets == {"v1": 2}

    def test_scale_up_cheapest(self):
        a = self._analysis(
            [
                rm("p0", 0.78, 4, variant="cheap", cost=5),
                rm("p1", 0.78, 4, variant="pricey", cost=50),
            ]
        )
        assert a.should_scale_up
        targets = AN.calculate_saturation_targets(
            a, [vs("cheap", 1), vs("pricey", 1)]
        )
        assert targets == {"cheap": 2, "pricey": 1}

    def test_scale_up_tie_break_alphabetical(self):
        a = self._analysis(
            [
                rm("p0", 0.78, 4, variant="bbb", cost=10),
                rm("p1", 0.78, 4, variant="aaa", cost=10),
            ]
        )
        targets = AN.calculate_saturation_targets(a, [vs("aaa", 1), vs("bbb", 1)])
        assert targets["aaa"] == 2 and targets["bbb"] == 1

    def test_scale_up_skips_pending(self):
        a = self._analysis(
            [
                rm("p0", 0.78, 4, variant="cheap", cost=5),
                rm("p1", 0.78, 4, variant="pricey", cost=50),
            ]
        )
        targets = AN.calculate_saturation_targets(
            a, [vs("cheap", 1, pending=1), vs("pricey", 1)]
        )
        # Transition check: metrics(1)==current(1) for both; pending blocks cheap
        assert targets == {"cheap": 1, "pricey": 2}

    def test_scale_down_most_expensive(self):
        a = self._analysis(
            [
                rm("p0", 0.05, 0, variant="cheap", cost=5),
                rm("p1", 0.05, 0, variant="cheap", cost=5),
                rm("p2", 0.05, 0, variant="pricey", cost=50),
                rm("p3", 0.05, 0, variant="pricey", cost=50),
            ]
        )
        assert a.scale_down_safe
        targets = AN.calculate_saturation_targets(
            a, [vs("cheap", 2), vs("pricey", 2)]
        )
        assert targets == {"cheap": 2, "pricey": 1}

    def test_scale_down_floor_one(self):
        a = self._analysis(
            [rm("p0", 0.05, 0, variant="only"), rm("p1", 0.05, 0, variant="only")]
        )
        # current=2, can go to 1
        targets = AN.calculate_saturation_targets(a, [vs("only", 2)])
        assert targets == {"only": 1}
        # but a variant at 1 replica can't scale down — need a stable analysis
        # with one ready replica that is still scale-down-safe is impossible
        # (needs >=2 non-saturated), covered by test_scale_down_requires_two

    def test_transition_freeze_desired_mismatch(self):
        a = self._analysis(
            [rm("p0", 0.78, 4, variant="v1"), rm("p1", 0.78, 4, variant="v1")]
        )
        assert a.should_scale_up
        # desired=3, current=2 → frozen at desired
        targets = AN.calculate_saturation_targets(a, [vs("v1", 2, desired=3)])
        assert targets == {"v1": 3}

    def test_transition_freeze_metrics_mismatch(self):
        a = self._analysis([rm("p0", 0.78, 4, variant="v1")])
        # metrics=1 but current=2 → frozen at current
        targets = AN.calculate_saturation_targets(a, [vs("v1", 2)])
        assert targets == {"v1": 2}

    def test_stable_base_is_metrics_count(self):
        a = self._analysis(
            [rm("p0", 0.78, 4, variant="v1"), rm("p1", 0.78, 4, variant="v1")]
        )
        targets = AN.calculate_saturation_targets(a, [vs("v1", 2)])
        assert targets == {"v1": 3}  # ready-replica base + 1
