"""Inferno library tests: queueing models, analyzer sizing, system/solver,
EKF tuner. Mirrors reference pkg/{analyzer,core,solver} test coverage
(queueanalyzer_test 1799, core 4487, greedy_test 2755 LoC).
"""
import math

import pytest

from wva_amd.inferno import (
    Configuration,
    Manager,
    MM1KModel,
    MM1StateDependentModel,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    Solver,
    System,
    TargetPerf,
    binary_search,
)
from wva_amd.inferno.types import (
    AcceleratorSpec,
    ModelAcceleratorPerfData,
    ModelTarget,
    OptimizerSpec,
    ServerLoadSpec,
    ServerSpec,
    ServiceClassSpec,
    ServiceParmsSpec,
    SystemData,
    POLICY_PRIORITY_EXHAUSTIVE,
)
from wva_amd.inferno.tuner import Observation, ServiceParmsTuner, TunerConfig


class TestMM1K:
    def test_light_load(self):
        m = MM1KModel(K=10)
        m.solve(lam=0.1, mu=1.0)
        assert m.is_valid
        # M/M/1 approx at rho=0.1: N ≈ rho/(1-rho) = 0.111
        assert abs(m.avg_num_in_system - 0.111) < 0.01
        assert abs(m.throughput - 0.1) < 1e-3

    def test_probabilities_sum_to_one(self):
        m = MM1KModel(K=5)
        m.solve(lam=0.8, mu=1.0)
        assert abs(sum(m.p) - 1.0) < 1e-9

    def test_rho_one(self):
        m = MM1KModel(K=4)
        m.solve(lam=1.0, mu=1.0)
        assert m.is_valid
        # uniform distribution at rho=1
        assert all(abs(p - 0.2) < 1e-9 for p in m.p)

    def test_saturated_throughput_capped(self):
        m = MM1KModel(K=3)
        m.solve(lam=2.0, mu=1.0)
        assert m.is_valid
        assert m.throughput < 1.0  # can't exceed service rate

    def test_invalid_inputs(self):
        m = MM1KModel(K=3)
        m.solve(lam=-1, mu=1)
        assert not m.is_valid
        m.solve(lam=1, mu=0)
        assert not m.is_valid


class TestStateDependent:
    def test_matches_mm1k_with_constant_rates(self):
        K = 8
        sd = MM1StateDependentModel(K, [1.0] * K)
        mm = MM1KModel(K)
        sd.solve(0.5)
        mm.solve(0.5, 1.0)
        assert abs(sd.avg_num_in_system - mm.avg_num_in_system) < 1e-6
        assert abs(sd.throughput - mm.throughput) < 1e-6

    def test_batch_speedup_increases_throughput(self):
        # service rate grows with occupancy (continuous batching)
        K = 16
        increasing = MM1StateDependentModel(K, [0.1 * n for n in range(1, 9)])
        flat = MM1StateDependentModel(K, [0.1] * 8)
        increasing.solve(0.5)
        flat.solve(0.5)
        assert increasing.throughput > flat.throughput

    def test_avg_in_servers_bounded(self):
        m = MM1StateDependentModel(20, [0.5] * 4)  # 4 "servers"
        m.solve(10.0)
        assert m.avg_num_in_servers <= 4.0 + 1e-9


class TestBinarySearch:
    def test_increasing(self):
        x, ind = binary_search(0, 10, 25.0, lambda x: x * x)
        assert ind == 0 and abs(x - 5.0) < 1e-3

    def test_decreasing(self):
        x, ind = binary_search(1, 10, 0.5, lambda x: 1.0 / x)
        assert ind == 0 and abs(x - 2.0) < 1e-3

    def test_below_region(self):
        x, ind = binary_search(1, 10, 0.5, lambda x: x)
        assert ind == -1 and x == 1

    def test_above_region(self):
        x, ind = binary_search(1, 10, 50, lambda x: x)
        assert ind == 1 and x == 10


def make_analyzer(alpha=20.58, beta=0.41, gamma=0.0, max_batch=4,
                  avg_in=100, avg_out=50):
    return QueueAnalyzer(
        Configuration(
            max_batch_size=max_batch,
            max_queue_size=max_batch * 10,
            service_parms=ServiceParms(alpha=alpha, beta=beta, gamma=gamma),
        ),
        RequestSize(avg_input_tokens=avg_in, avg_output_tokens=avg_out),
    )


class TestQueueAnalyzer:
    def test_service_rate_shapes(self):
        qa = make_analyzer()
        assert len(qa.serv_rate) == 4
        # higher batch → higher aggregate service rate for small alpha share
        assert qa.serv_rate[-1] > qa.serv_rate[0]

    def test_analyze_monotone_in_rate(self):
        qa = make_analyzer()
        lo = qa.analyze(qa.rate_min * 2)
        hi = qa.analyze(qa.rate_max * 0.9)
        assert hi.avg_ttft > lo.avg_ttft
        assert hi.avg_token_time >= lo.avg_token_time
        assert hi.rho > lo.rho

    def test_reference_shaped_sizing(self):
        """Reference docs/integrations/hpa-integration.md:190-193 worked
        example shape (α=20.58, β=0.41, maxBatch=4, ITL 24 ms / TTFT-wait
        500 ms). The doc's exact 25.31 req/min figure depends on the
        emulated workload's avg token counts, which are not published, so
        we assert internal consistency: sizing respects both targets and
        the sized operating point is stable (ρ < 1)."""
        qa = make_analyzer()
        rates, metrics, achieved = qa.size(
            TargetPerf(target_ttft=500.0, target_itl=24.0)
        )
        assert achieved.target_itl <= 24.0 * 1.05
        assert achieved.target_ttft <= 500.0 * 1.05
        assert 0 < metrics.rho <= 1.0
        assert metrics.throughput <= qa.rate_max

    def test_size_itl_binding(self):
        qa = make_analyzer()
        rates, metrics, achieved = qa.size(TargetPerf(target_itl=24.0))
        assert achieved.target_itl <= 24.0 * 1.05
        assert rates.rate_target_itl <= qa.rate_max

    def test_size_tps_stability_margin(self):
        qa = make_analyzer()
        rates, _, _ = qa.size(TargetPerf(target_tps=100.0))
        assert abs(rates.rate_target_tps - qa.rate_max * 0.9) < 1e-6

    def test_invalid_rate(self):
        qa = make_analyzer()
        with pytest.raises(ValueError):
            qa.analyze(qa.rate_max * 2)
        with pytest.raises(ValueError):
            qa.analyze(0)


def mi355x_system(capacity=16, arrival_rpm=40.0):
    """One model on two accelerator options (MI355X vs MI300X)."""
    data = SystemData(
        accelerators=[
            AcceleratorSpec(name="MI355X", type="MI355X", cost=50.0),
            AcceleratorSpec(name="MI300X", type="MI300X", cost=30.0),
        ],
        models=[
            ModelAcceleratorPerfData(
                name="llama-8b", acc="MI355X", acc_count=1,
                max_batch_size=256, at_tokens=50,
                service_parms=ServiceParmsSpec(alpha=11.28, beta=0.0152),
            ),
            ModelAcceleratorPerfData(
                name="llama-8b", acc="MI300X", acc_count=1,
                max_batch_size=256, at_tokens=50,
                service_parms=ServiceParmsSpec(alpha=18.0, beta=0.05),
            ),
        ],
        service_classes=[
            ServiceClassSpec(
                name="premium", priority=1,
                model_targets=[
                    ModelTarget(model="llama-8b", slo_itl=24.0, slo_ttft=500.0)
                ],
            )
        ],
        servers=[
            ServerSpec(
                name="srv-a", service_class="premium", model="llama-8b",
                load=ServerLoadSpec(
                    arrival_rate=arrival_rpm, avg_in_tokens=100,
                    avg_out_tokens=50,
                ),
            )
        ],
        capacity={"MI355X": capacity, "MI300X": capacity},
    )
    return System(data)


class TestSystemSolver:
    def test_allocation_created(self):
        sys_ = mi355x_system()
        sys_.generate_all_allocations()
        srv = sys_.servers["srv-a"]
        assert "MI355X" in srv.all_allocations
        alloc = srv.all_allocations["MI355X"]
        assert alloc.num_replicas >= 1
        assert alloc.cost == 50.0 * alloc.num_replicas
        assert alloc.itl <= 24.0 * 1.05

    def test_unlimited_picks_min_value(self):
        sys_ = mi355x_system()
        mgr = Manager(sys_, OptimizerSpec(unlimited=True))
        mgr.optimize()
        srv = sys_.servers["srv-a"]
        assert srv.allocation is not None
        values = [a.value for a in srv.all_allocations.values()]
        assert srv.allocation.value == min(values)

    def test_greedy_respects_capacity(self):
        # capacity 0 on the cheap option forces the expensive one
        sys_ = mi355x_system()
        sys_.capacity = {"MI355X": 16, "MI300X": 0}
        mgr = Manager(sys_, OptimizerSpec())
        mgr.optimize()
        srv = sys_.servers["srv-a"]
        assert srv.allocation is not None
        assert srv.allocation.accelerator == "MI355X"

    def test_greedy_no_capacity_no_allocation(self):
        sys_ = mi355x_system()
        sys_.capacity = {"MI355X": 0, "MI300X": 0}
        mgr = Manager(sys_, OptimizerSpec())
        mgr.optimize()
        assert sys_.servers["srv-a"].allocation is None

    def test_best_effort_partial(self):
        sys_ = mi355x_system(arrival_rpm=40000.0)  # needs many replicas
        sys_.generate_all_allocations()
        want = min(
            a.num_replicas
            for a in sys_.servers["srv-a"].all_allocations.values()
        )
        sys_.capacity = {"MI355X": 1, "MI300X": 1}
        assert want > 1
        solver = Solver(OptimizerSpec(
            saturation_policy=POLICY_PRIORITY_EXHAUSTIVE
        ))
        solver.solve(sys_)
        srv = sys_.servers["srv-a"]
        assert srv.allocation is not None
        assert srv.allocation.num_replicas == 1  # partial best-effort

    def test_priority_ordering(self):
        """High-priority (lower value) server gets scarce capacity first."""
        sys_ = mi355x_system()
        sys_.service_classes["besteffort"] = ServiceClassSpec(
            name="besteffort", priority=10,
            model_targets=[
                ModelTarget(model="llama-8b", slo_itl=24.0, slo_ttft=500.0)
            ],
        )
        sys_.add_server(ServerSpec(
            name="srv-b", service_class="besteffort", model="llama-8b",
            load=ServerLoadSpec(
                arrival_rate=40.0, avg_in_tokens=100, avg_out_tokens=50
            ),
        ))
        sys_.generate_all_allocations()
        # capacity for exactly one replica of one server on MI300X (cheaper)
        sys_.capacity = {"MI355X": 0, "MI300X": 1}
        solver = Solver(OptimizerSpec())
        solver.solve(sys_)
        assert sys_.servers["srv-a"].allocation is not None
        assert sys_.servers["srv-b"].allocation is None

    def test_transition_penalty(self):
        sys_ = mi355x_system()
        sys_.servers["srv-a"].spec.current_accelerator = "MI355X"
        sys_.generate_all_allocations()
        a355 = sys_.servers["srv-a"].all_allocations["MI355X"]
        a300 = sys_.servers["srv-a"].all_allocations["MI300X"]
        assert a355.value == a355.cost  # no penalty, same accelerator
        assert abs(a300.value - a300.cost * 1.1) < 1e-6  # switch factor 0.1

    def test_diff_allocation(self):
        sys_ = mi355x_system()
        mgr = Manager(sys_, OptimizerSpec(unlimited=True))
        diff = mgr.optimize()
        assert "srv-a" in diff
        assert diff["srv-a"].replicas_from == 0
        assert diff["srv-a"].replicas_to >= 1
        # second run: no change
        diff2 = mgr.optimize()
        assert "srv-a" not in diff2


class TestTuner:
    def _observe(self, true_parms, rate, n=1):
        """Generate synthetic observations from a ground-truth model."""
        qa = QueueAnalyzer(
            Configuration(
                max_batch_size=256, max_queue_size=2560,
                service_parms=true_parms,
            ),
            RequestSize(avg_input_tokens=100, avg_output_tokens=50),
        )
        m = qa.analyze(rate)
        return Observation(
            request_rate=rate, avg_input_tokens=100, avg_output_tokens=50,
            ttft_ms=m.avg_ttft, itl_ms=m.avg_token_time,
        )

    def test_converges_toward_truth(self):
        true = ServiceParms(alpha=11.0, beta=0.015, gamma=0.0)
        tuner = ServiceParmsTuner(
            ServiceParms(alpha=20.0, beta=0.05, gamma=0.0),
            TunerConfig(measurement_noise=0.01),
        )
        err0 = abs(tuner.parms().alpha - true.alpha)
        for rate in [50, 100, 200, 150, 80, 120, 60, 180] * 3:
            tuner.update(self._observe(true, rate))
        err1 = abs(tuner.parms().alpha - true.alpha)
        assert err1 < err0  # moving toward ground truth

    def test_outlier_rejected(self):
        true = ServiceParms(alpha=11.0, beta=0.015)
        tuner = ServiceParmsTuner(
            ServiceParms(alpha=11.0, beta=0.015),
            TunerConfig(measurement_noise=0.01),
        )
        ok = tuner.update(self._observe(true, 100))
        assert ok
        outlier = Observation(
            request_rate=100, avg_input_tokens=100, avg_output_tokens=50,
            ttft_ms=50000.0, itl_ms=5000.0,
        )
        before = tuner.parms()
        assert not tuner.update(outlier)
        after = tuner.parms()
        assert before.alpha == after.alpha  # state untouched by outlier

    def test_rollback_after_consecutive_rejections(self):
        true = ServiceParms(alpha=11.0, beta=0.015)
        tuner = ServiceParmsTuner(
            ServiceParms(alpha=11.0, beta=0.015),
            TunerConfig(measurement_noise=0.01, max_consecutive_rejections=3),
        )
        outlier = Observation(
            request_rate=100, avg_input_tokens=100, avg_output_tokens=50,
            ttft_ms=50000.0, itl_ms=5000.0,
        )
        for _ in range(3):
            tuner.update(outlier)
        # stash consumed, covariance widened — next good update accepted
        assert tuner._stash is None
        assert tuner.update(self._observe(true, 100))


class TestServiceClassConfigMap:
    def test_reference_chart_format_loads(self):
        """Byte-format parity with the reference chart's service-class
        ConfigMap (slo-tpot/slo-ttft keys, priority, per-model data)."""
        from wva_amd.inferno.types import parse_service_class_configmap

        data = {
            "premium.yaml": (
                "name: Premium\n"
                "priority: 1\n"
                "data:\n"
                "  - model: default/default\n"
                "    slo-tpot: 24\n"
                "    slo-ttft: 500\n"
                "  - model: meta/llama0-70b\n"
                "    slo-tpot: 80\n"
                "    slo-ttft: 500\n"
            ),
            "freemium.yaml": (
                "name: Freemium\n"
                "priority: 10\n"
                "data:\n"
                "  - model: ibm/granite-13b\n"
                "    slo-tpot: 200\n"
                "    slo-ttft: 2000\n"
            ),
        }
        classes = parse_service_class_configmap(data)
        by_name = {c.name: c for c in classes}
        assert by_name["Premium"].priority == 1
        assert by_name["Freemium"].priority == 10
        t = by_name["Premium"].model_targets[0]
        assert t.model == "default/default"
        assert t.slo_itl == 24.0
        assert t.slo_ttft == 500.0


class TestCapacitySpec:
    def test_wire_shape(self):
        from wva_amd.inferno.types import CapacitySpec

        spec = CapacitySpec.from_dict({"count": {"MI355X": 16, "MI300X": "8"}})
        assert spec.counts == {"MI355X": 16, "MI300X": 8}

    def test_bare_map_and_garbage(self):
        from wva_amd.inferno.types import CapacitySpec

        assert CapacitySpec.from_dict({"MI355X": 4}).counts == {"MI355X": 4}
        assert CapacitySpec.from_dict(None).counts == {}
        assert CapacitySpec.from_dict({"count": "x"}).counts == {}
        assert CapacitySpec.from_dict(
            {"count": {"MI355X": "not-a-number"}}
        ).counts == {}

    def test_system_data_uses_it(self):
        data = SystemData.from_dict({
            "spec": {"capacity": {"count": {"MI355X": 12}}},
        })
        assert data.capacity == {"MI355X": 12}


class TestModelAnalyzer:
    """Reference pkg/analyzer ModelAnalyzer parity: per-model feasible
    allocations across every accelerator, independent of the solver."""

    def _va(self):
        from wva_amd.api.types import (
            ObjectMeta, VariantAutoscaling, VariantAutoscalingSpec,
        )

        return VariantAutoscaling(
            metadata=ObjectMeta(name="srv-a", namespace="default"),
            spec=VariantAutoscalingSpec(model_id="llama-8b"),
        )

    def test_feasible_on_both_accelerators(self):
        from wva_amd.analyzers.modelanalyzer import ModelAnalyzer

        sys_ = mi355x_system()
        out = ModelAnalyzer(sys_).analyze_model(
            self._va(), arrival_rate_per_min=40.0,
            avg_in_tokens=100, avg_out_tokens=50,
            service_class="premium",
        )
        assert set(out) == {"MI355X", "MI300X"}
        for alloc in out.values():
            assert alloc.num_replicas >= 1
            assert alloc.itl <= 24.0 * 1.05  # meets the premium SLO

    def test_infeasible_accelerator_omitted(self):
        """An accelerator whose floor ITL (α+β at batch 1) already breaks
        the SLO yields no allocation for that accelerator."""
        from wva_amd.analyzers.modelanalyzer import ModelAnalyzer

        data = SystemData(
            accelerators=[
                AcceleratorSpec(name="MI355X", type="MI355X", cost=50.0),
                AcceleratorSpec(name="SLOW", type="SLOW", cost=1.0),
            ],
            models=[
                ModelAcceleratorPerfData(
                    name="llama-8b", acc="MI355X", acc_count=1,
                    max_batch_size=256, at_tokens=50,
                    service_parms=ServiceParmsSpec(alpha=11.28, beta=0.0152),
                ),
                ModelAcceleratorPerfData(
                    name="llama-8b", acc="SLOW", acc_count=1,
                    max_batch_size=256, at_tokens=50,
                    service_parms=ServiceParmsSpec(alpha=100.0, beta=5.0),
                ),
            ],
            service_classes=[
                ServiceClassSpec(
                    name="premium", priority=1,
                    model_targets=[ModelTarget(
                        model="llama-8b", slo_itl=24.0, slo_ttft=500.0,
                    )],
                )
            ],
            capacity={"MI355X": 16, "SLOW": 16},
        )
        from wva_amd.analyzers.modelanalyzer import ModelAnalyzer as MA

        out = MA(System(data)).analyze_model(
            self._va(), arrival_rate_per_min=40.0,
            avg_in_tokens=100, avg_out_tokens=50,
            service_class="premium",
        )
        assert "MI355X" in out
        assert "SLOW" not in out

    def test_higher_load_needs_more_replicas(self):
        from wva_amd.analyzers.modelanalyzer import ModelAnalyzer

        lo = ModelAnalyzer(mi355x_system()).analyze_model(
            self._va(), 40.0, 100, 50, service_class="premium",
        )["MI355X"].num_replicas
        hi = ModelAnalyzer(mi355x_system()).analyze_model(
            self._va(), 100_000.0, 100, 50, service_class="premium",
        )["MI355X"].num_replicas
        assert hi > lo
