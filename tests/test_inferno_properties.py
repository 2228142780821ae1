"""Property-based tests for the Inferno queueing library.

The reference's heaviest unit suites live here (pkg/solver/greedy_test.go
2755 LoC, pkg/analyzer/queueanalyzer_test.go 1799, pkg/core 4487):
instead of porting their enumerated tables, hypothesis generates the
state space and asserts the INVARIANTS those tables spot-check —
queueing-theory identities (Little's law, monotonicity, throughput
bounds), sizing inverses (size→analyze round-trip meets the SLO), and
solver feasibility/priority properties.
"""
import pytest
from hypothesis import assume, given, settings, strategies as st

from wva_amd.inferno.manager import Manager
from wva_amd.inferno.queue_analyzer import (
    Configuration,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
    TargetPerf,
)
from wva_amd.inferno.queueing import MM1KModel, MM1StateDependentModel
from wva_amd.inferno.system import System
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
)

RATES = st.floats(min_value=0.01, max_value=5.0)
SERVICE = st.floats(min_value=0.1, max_value=10.0)
K_SIZES = st.integers(min_value=2, max_value=200)


class TestMM1KProperties:
    @given(lam=RATES, mu=SERVICE, K=K_SIZES)
    @settings(max_examples=200, deadline=None)
    def test_probabilities_normalized_and_nonnegative(self, lam, mu, K):
        m = MM1KModel(K)
        m.solve(lam, mu)
        assume(m.is_valid)
        assert all(p >= -1e-12 for p in m.p)
        assert sum(m.p) == pytest.approx(1.0, abs=1e-9)

    @given(lam=RATES, mu=SERVICE, K=K_SIZES)
    @settings(max_examples=200, deadline=None)
    def test_littles_law(self, lam, mu, K):
        """N = λ_eff · T — the identity every queueing result must
        satisfy (mm1kmodel.go Little's-law response time)."""
        m = MM1KModel(K)
        m.solve(lam, mu)
        assume(m.is_valid and m.throughput > 1e-12)
        assert m.avg_num_in_system == pytest.approx(
            m.throughput * m.avg_resp_time, rel=1e-6
        )

    @given(lam=RATES, mu=SERVICE, K=K_SIZES)
    @settings(max_examples=200, deadline=None)
    def test_throughput_bounded_by_offer_and_service(self, lam, mu, K):
        m = MM1KModel(K)
        m.solve(lam, mu)
        assume(m.is_valid)
        assert 0 <= m.throughput <= min(lam, mu) * (1 + 1e-9)

    @given(lam=RATES, mu=SERVICE, K=K_SIZES)
    @settings(max_examples=100, deadline=None)
    def test_larger_buffer_never_reduces_throughput(self, lam, mu, K):
        m1, m2 = MM1KModel(K), MM1KModel(K + 10)
        m1.solve(lam, mu)
        m2.solve(lam, mu)
        assume(m1.is_valid and m2.is_valid)
        assert m2.throughput >= m1.throughput - 1e-9

    @given(lam=RATES, mu=SERVICE, K=st.integers(min_value=2, max_value=64))
    @settings(max_examples=100, deadline=None)
    def test_state_dependent_reduces_to_mm1k(self, lam, mu, K):
        """Constant service rates ⇒ the state-dependent model IS
        M/M/1/K (mm1modelstatedependent.go:9-128 invariant)."""
        sd = MM1StateDependentModel(K, [mu] * K)
        sd.solve(lam)
        mm = MM1KModel(K)
        mm.solve(lam, mu)
        assume(mm.is_valid and sd.is_valid)
        assert sd.throughput == pytest.approx(mm.throughput, rel=1e-6)
        assert sd.avg_num_in_system == pytest.approx(
            mm.avg_num_in_system, rel=1e-6
        )

    @given(lam=RATES, base=SERVICE,
           K=st.integers(min_value=4, max_value=64))
    @settings(max_examples=100, deadline=None)
    def test_batching_speedup_never_hurts(self, lam, base, K):
        """Occupancy-growing service rates (continuous batching) never
        yield worse throughput than the constant-rate floor."""
        grow = MM1StateDependentModel(
            K, [base * max(1, n) ** 0.5 for n in range(1, K + 1)]
        )
        flat = MM1StateDependentModel(K, [base] * K)
        grow.solve(lam)
        flat.solve(lam)
        assume(grow.is_valid and flat.is_valid)
        assert grow.throughput >= flat.throughput - 1e-9

    @given(lam=RATES, mu=SERVICE, K=K_SIZES)
    @settings(max_examples=100, deadline=None)
    def test_state_probabilities_littles_law_sd(self, lam, mu, K):
        sd = MM1StateDependentModel(K, [mu] * min(K, 8))
        sd.solve(lam)
        assume(sd.is_valid and sd.throughput > 1e-12)
        assert sd.avg_num_in_system == pytest.approx(
            sd.throughput * sd.avg_resp_time, rel=1e-6
        )
        # servers-occupancy never exceeds the number of service slots
        assert sd.avg_num_in_servers <= min(K, 8) + 1e-9


ALPHAS = st.floats(min_value=2.0, max_value=100.0)
BETAS = st.floats(min_value=0.01, max_value=2.0)
TOKENS = st.integers(min_value=10, max_value=500)
BATCHES = st.integers(min_value=2, max_value=128)


def _qa(alpha, beta, avg_in, avg_out, max_batch):
    return QueueAnalyzer(
        Configuration(
            max_batch_size=max_batch,
            max_queue_size=max_batch * 10,
            service_parms=ServiceParms(alpha=alpha, beta=beta),
        ),
        RequestSize(
            avg_input_tokens=float(avg_in), avg_output_tokens=float(avg_out)
        ),
    )


class TestQueueAnalyzerProperties:
    @given(alpha=ALPHAS, beta=BETAS, avg_in=TOKENS, avg_out=TOKENS,
           max_batch=BATCHES)
    @settings(max_examples=150, deadline=None)
    def test_latency_monotone_in_rate(self, alpha, beta, avg_in, avg_out,
                                      max_batch):
        """TTFT/ITL non-decreasing in arrival rate — the monotonicity
        the binary-search sizing depends on (queueanalyzer.go:181-258)."""
        qa = _qa(alpha, beta, avg_in, avg_out, max_batch)
        lo = qa.rate_min * 1.05
        hi = qa.rate_max * 0.95
        assume(hi > lo * 1.2)
        mid = (lo + hi) / 2
        m_lo, m_mid, m_hi = qa.analyze(lo), qa.analyze(mid), qa.analyze(hi)
        eps = 1e-6
        assert m_lo.avg_ttft <= m_mid.avg_ttft * (1 + eps)
        assert m_mid.avg_ttft <= m_hi.avg_ttft * (1 + eps)
        assert m_lo.avg_token_time <= m_mid.avg_token_time * (1 + eps)
        assert m_mid.avg_token_time <= m_hi.avg_token_time * (1 + eps)
        assert m_lo.rho <= m_mid.rho <= m_hi.rho * (1 + eps)

    @given(alpha=ALPHAS, beta=BETAS, avg_in=TOKENS, avg_out=TOKENS,
           max_batch=BATCHES,
           slack=st.floats(min_value=1.5, max_value=10.0))
    @settings(max_examples=150, deadline=None)
    def test_size_then_analyze_meets_slo(self, alpha, beta, avg_in,
                                         avg_out, max_batch, slack):
        """Sizing inverse: the rate size() returns must, when analyzed,
        satisfy the SLO it was sized for (within the search tolerance)."""
        qa = _qa(alpha, beta, avg_in, avg_out, max_batch)
        base = qa.analyze(qa.rate_min * 1.05)
        target = TargetPerf(
            target_ttft=base.avg_ttft * slack,
            target_itl=base.avg_token_time * slack,
        )
        try:
            _rates, metrics, achieved = qa.size(target)
        except ValueError:
            assume(False)
            return
        assert achieved.target_ttft <= target.target_ttft * 1.05
        assert achieved.target_itl <= target.target_itl * 1.05
        assert 0 <= metrics.rho <= 1.0 + 1e-9
        assert metrics.throughput <= qa.rate_max * (1 + 1e-9)

    @given(alpha=ALPHAS, beta=BETAS, avg_out=TOKENS, max_batch=BATCHES)
    @settings(max_examples=100, deadline=None)
    def test_itl_floor_is_service_curve(self, alpha, beta, avg_out,
                                        max_batch):
        """ITL can never beat the batch-1 iteration time α+β."""
        qa = _qa(alpha, beta, 100, avg_out, max_batch)
        m = qa.analyze(qa.rate_min * 1.05)
        assert m.avg_token_time >= (alpha + beta) * 0.99

    @given(alpha=ALPHAS, beta=BETAS, avg_in=TOKENS, avg_out=TOKENS,
           max_batch=BATCHES)
    @settings(max_examples=100, deadline=None)
    def test_rate_max_is_actually_analyzable(self, alpha, beta, avg_in,
                                             avg_out, max_batch):
        qa = _qa(alpha, beta, avg_in, avg_out, max_batch)
        m = qa.analyze(qa.rate_max)
        assert m.throughput > 0
        with pytest.raises(ValueError):
            qa.analyze(qa.rate_max * 1.01)


def _solver_system(costs, capacity, rates, slo_itl=24.0):
    """N servers (one per rate) on the given accelerator cost map."""
    data = SystemData(
        accelerators=[
            AcceleratorSpec(name=n, type=n, cost=c) for n, c in costs.items()
        ],
        models=[
            ModelAcceleratorPerfData(
                name="m", acc=n, acc_count=1, max_batch_size=256,
                at_tokens=50,
                service_parms=ServiceParmsSpec(alpha=11.28, beta=0.0152),
            )
            for n in costs
        ],
        service_classes=[ServiceClassSpec(
            name="premium", priority=1,
            model_targets=[ModelTarget(model="m", slo_itl=slo_itl,
                                       slo_ttft=500.0)],
        )],
        servers=[
            ServerSpec(
                name=f"srv{i}", service_class="premium", model="m",
                load=ServerLoadSpec(arrival_rate=r, avg_in_tokens=100,
                                    avg_out_tokens=50),
            )
            for i, r in enumerate(rates)
        ],
        capacity=dict(capacity),
    )
    return System(data)


class TestSolverProperties:
    @given(cap=st.integers(min_value=0, max_value=32),
           rates=st.lists(st.floats(min_value=10.0, max_value=5000.0),
                          min_size=1, max_size=4))
    @settings(max_examples=60, deadline=None)
    def test_greedy_never_exceeds_capacity(self, cap, rates):
        system = _solver_system({"MI355X": 50.0}, {"MI355X": cap}, rates)
        Manager(system, OptimizerSpec(unlimited=False)).optimize()
        used = sum(
            srv.allocation.num_replicas
            * system.units_per_replica("m", srv.allocation.accelerator)
            for srv in system.servers.values()
            if srv.allocation is not None
        )
        assert used <= cap

    @given(rate=st.floats(min_value=10.0, max_value=2000.0))
    @settings(max_examples=40, deadline=None)
    def test_unlimited_picks_min_value_allocation(self, rate):
        system = _solver_system(
            {"CHEAP": 10.0, "PRICY": 90.0},
            {"CHEAP": 10_000, "PRICY": 10_000}, [rate],
        )
        Manager(system, OptimizerSpec(unlimited=True)).optimize()
        srv = system.servers["srv0"]
        assume(srv.allocation is not None)
        values = [a.value for a in srv.all_allocations.values()]
        assert srv.allocation.value == min(values)

    @given(cap=st.integers(min_value=0, max_value=8),
           rate=st.floats(min_value=10.0, max_value=2000.0))
    @settings(max_examples=60, deadline=None)
    def test_scarcity_serves_high_priority_first(self, cap, rate):
        """Priority-1 ahead of priority-10 under scarcity
        (greedy.go:35-104 priority groups): if the low-priority server
        got replicas, the high-priority one is already served."""
        data = SystemData(
            accelerators=[AcceleratorSpec(name="MI355X", type="MI355X",
                                          cost=50.0)],
            models=[ModelAcceleratorPerfData(
                name="m", acc="MI355X", acc_count=1, max_batch_size=256,
                at_tokens=50,
                service_parms=ServiceParmsSpec(alpha=11.28, beta=0.0152),
            )],
            service_classes=[
                ServiceClassSpec(name="prem", priority=1, model_targets=[
                    ModelTarget(model="m", slo_itl=24.0, slo_ttft=500.0)]),
                ServiceClassSpec(name="free", priority=10, model_targets=[
                    ModelTarget(model="m", slo_itl=24.0, slo_ttft=500.0)]),
            ],
            servers=[
                ServerSpec(name="gold", service_class="prem", model="m",
                           load=ServerLoadSpec(arrival_rate=rate,
                                               avg_in_tokens=100,
                                               avg_out_tokens=50)),
                ServerSpec(name="bronze", service_class="free", model="m",
                           load=ServerLoadSpec(arrival_rate=rate,
                                               avg_in_tokens=100,
                                               avg_out_tokens=50)),
            ],
            capacity={"MI355X": cap},
        )
        system = System(data)
        Manager(system, OptimizerSpec(unlimited=False)).optimize()
        gold = system.servers["gold"].allocation
        bronze = system.servers["bronze"].allocation
        if bronze is not None and bronze.num_replicas > 0:
            assert gold is not None and gold.num_replicas > 0

    @given(rate=st.floats(min_value=10.0, max_value=2000.0),
           cap=st.integers(min_value=1, max_value=64))
    @settings(max_examples=60, deadline=None)
    def test_solution_allocations_are_feasible_candidates(self, rate, cap):
        """Whatever the solver picks must be one of the server's own
        generated candidate allocations (never a fabricated one)."""
        system = _solver_system({"MI355X": 50.0}, {"MI355X": cap}, [rate])
        Manager(system, OptimizerSpec(unlimited=False)).optimize()
        srv = system.servers["srv0"]
        if srv.allocation is not None:
            cand = srv.all_allocations.get(srv.allocation.accelerator)
            assert cand is not None
            assert srv.allocation.num_replicas <= cand.num_replicas


class TestV1AnalyzerProperties:
    """Property sweep over the V1 percentage analyzer (reference
    internal/saturation/analyzer.go:31-280): threshold monotonicity and
    scale-down safety invariants across generated replica states."""

    def _analyze(self, kvs_qs):
        from wva_amd.analyzers.interfaces import ReplicaMetrics
        from wva_amd.analyzers.saturation_v1 import SaturationAnalyzerV1
        from wva_amd.config.saturation import SaturationScalingConfig

        metrics = [
            ReplicaMetrics(pod_name=f"p{i}", kv_cache_usage=kv,
                           queue_length=q, variant_name="v",
                           accelerator_name="MI355X", cost=10.0)
            for i, (kv, q) in enumerate(kvs_qs)
        ]
        return SaturationAnalyzerV1().analyze_model_saturation(
            "m", "ns", metrics, SaturationScalingConfig()
        )

    @given(st.lists(
        st.tuples(st.floats(min_value=0, max_value=1),
                  st.integers(min_value=0, max_value=20)),
        min_size=1, max_size=8,
    ))
    @settings(max_examples=200, deadline=None)
    def test_saturation_classification_consistent(self, kvs_qs):
        """A replica is saturated iff kv >= 0.80 OR q >= 5 (defaults);
        the analysis counts must agree with the definition."""
        a = self._analyze(kvs_qs)
        expected_sat = sum(
            1 for kv, q in kvs_qs if kv >= 0.80 or q >= 5
        )
        assert a.total_replicas == len(kvs_qs)
        assert a.total_replicas - a.non_saturated_count == expected_sat

    @given(st.lists(
        st.tuples(st.floats(min_value=0, max_value=0.79),
                  st.integers(min_value=0, max_value=4)),
        min_size=2, max_size=8,
    ))
    @settings(max_examples=200, deadline=None)
    def test_scale_down_safety_monotone_in_load(self, kvs_qs):
        """If scale-down is UNSAFE at some load, it stays unsafe when
        every replica's load increases (monotonicity of the N/(N−1)
        redistribution simulation, analyzer.go:233-280)."""
        a_low = self._analyze(kvs_qs)
        heavier = [
            (min(kv + 0.1, 0.79), min(q + 1, 4)) for kv, q in kvs_qs
        ]
        a_high = self._analyze(heavier)
        if not a_low.scale_down_safe:
            assert not a_high.scale_down_safe

    @given(st.lists(
        st.tuples(st.floats(min_value=0, max_value=1),
                  st.integers(min_value=0, max_value=20)),
        min_size=1, max_size=8,
    ))
    @settings(max_examples=200, deadline=None)
    def test_never_up_and_down_together(self, kvs_qs):
        a = self._analyze(kvs_qs)
        assert not (a.should_scale_up and a.scale_down_safe)

    @given(st.integers(min_value=1, max_value=8))
    @settings(max_examples=50, deadline=None)
    def test_idle_fleet_scale_down_safe_when_big_enough(self, n):
        """All-idle replicas: scale-down safe iff >= 2 non-saturated
        (MinNonSaturatedReplicasForScaleDown, constants.go:8)."""
        a = self._analyze([(0.0, 0)] * n)
        assert a.scale_down_safe == (n >= 2)
        assert not a.should_scale_up


class TestTunerRobustness:
    """EKF tuner structural invariants under arbitrary observation
    streams (the reference tuner is dormant and untested against hostile
    inputs; here it is live on the engine path, so it must never emit
    NaN/negative parameters no matter what the collector feeds it)."""

    def _tuner(self, alpha, beta, gamma):
        from wva_amd.inferno.tuner import ServiceParmsTuner, TunerConfig

        return ServiceParmsTuner(
            ServiceParms(alpha=alpha, beta=beta, gamma=gamma),
            TunerConfig(max_batch_size=32, max_queue_size=128),
        )

    observations = st.lists(
        st.tuples(
            st.floats(min_value=0.0, max_value=1e4),    # request_rate
            st.floats(min_value=1.0, max_value=4096.0),  # avg_input
            st.floats(min_value=1.0, max_value=1024.0),  # avg_output
            st.floats(min_value=0.0, max_value=1e6),     # ttft_ms
            st.floats(min_value=0.0, max_value=1e5),     # itl_ms
            st.booleans(),                               # itl_only
        ),
        min_size=1,
        max_size=8,
    )

    @settings(max_examples=40, deadline=None)
    @given(
        seed=st.tuples(
            st.floats(min_value=0.5, max_value=100.0),   # alpha
            st.floats(min_value=1e-4, max_value=1.0),    # beta
            st.floats(min_value=0.0, max_value=1e-3),    # gamma
        ),
        stream=observations,
    )
    def test_theta_always_finite_and_physical(self, seed, stream):
        import numpy as np
        from wva_amd.inferno.tuner import Observation

        tuner = self._tuner(*seed)
        for rate, avg_in, avg_out, ttft, itl, itl_only in stream:
            tuner.update(
                Observation(
                    request_rate=rate,
                    avg_input_tokens=avg_in,
                    avg_output_tokens=avg_out,
                    ttft_ms=ttft,
                    itl_ms=itl,
                ),
                itl_only=itl_only,
            )
            assert np.all(np.isfinite(tuner.theta))
            assert np.all(np.isfinite(tuner.P))
            p = tuner.parms()
            assert p.alpha >= 1e-6
            assert p.beta >= 0.0
            assert p.gamma >= 0.0
            # covariance stays symmetric with non-negative diagonal
            assert np.allclose(tuner.P, tuner.P.T, atol=1e-6)
            assert np.all(np.diag(tuner.P) >= -1e-9)

    @settings(max_examples=30, deadline=None)
    @given(
        seed=st.tuples(
            st.floats(min_value=1.0, max_value=50.0),
            st.floats(min_value=1e-3, max_value=0.5),
            st.floats(min_value=0.0, max_value=1e-4),
        ),
        stream=observations,
    )
    def test_accepted_step_respects_trust_region(self, seed, stream):
        import numpy as np
        from wva_amd.inferno.tuner import Observation

        tuner = self._tuner(*seed)
        frac = tuner.config.max_step_frac
        for rate, avg_in, avg_out, ttft, itl, itl_only in stream:
            before = tuner.theta.copy()
            accepted = tuner.update(
                Observation(
                    request_rate=rate,
                    avg_input_tokens=avg_in,
                    avg_output_tokens=avg_out,
                    ttft_ms=ttft,
                    itl_ms=itl,
                ),
                itl_only=itl_only,
            )
            if accepted:
                limit = np.maximum(np.abs(before) * frac, [0.5, 1e-3, 1e-3])
                # +tiny: the non-negativity projection can only shrink
                assert np.all(np.abs(tuner.theta - before) <= limit + 1e-12)
            else:
                assert np.array_equal(tuner.theta, before)

    @settings(max_examples=20, deadline=None)
    @given(
        seed=st.tuples(
            st.floats(min_value=1.0, max_value=50.0),
            st.floats(min_value=1e-3, max_value=0.5),
            st.floats(min_value=0.0, max_value=1e-4),
        ),
    )
    def test_model_consistent_observation_accepted_and_stable(self, seed):
        """Feeding the tuner its OWN prediction is a zero-innovation
        observation: always accepted, θ unchanged."""
        import numpy as np
        from wva_amd.inferno.tuner import Observation

        tuner = self._tuner(*seed)
        obs = Observation(
            request_rate=1.0, avg_input_tokens=100, avg_output_tokens=50,
            ttft_ms=0.0, itl_ms=0.0,
        )
        pred = tuner._h(tuner.theta, obs)
        obs.ttft_ms, obs.itl_ms = float(pred[0]), float(pred[1])
        before = tuner.theta.copy()
        assert tuner.update(obs) is True
        assert np.allclose(tuner.theta, before, rtol=1e-6, atol=1e-9)


class TestV1TargetCalculationProperties:
    """V1 CalculateSaturationTargets invariants (analyzer.go:290-439):
    targets move by at most ±1 replica per tick, across ONE variant;
    any transitioning variant freezes the whole model; scale-up picks
    the cheapest candidate, scale-down the most expensive."""

    def _targets(self, rows, should_up=False, down_safe=False):
        from wva_amd.analyzers.interfaces import (
            ModelSaturationAnalysis,
            VariantReplicaState,
            VariantSaturationAnalysis,
        )
        from wva_amd.analyzers.saturation_v1 import SaturationAnalyzerV1

        analyses, states = [], []
        for i, (cost, cur, desired, pending) in enumerate(rows):
            name = f"v{i}"
            analyses.append(VariantSaturationAnalysis(
                variant_name=name, cost=cost, replica_count=cur,
                non_saturated_count=cur,
            ))
            states.append(VariantReplicaState(
                variant_name=name, current_replicas=cur,
                desired_replicas=desired, pending_replicas=pending,
            ))
        analysis = ModelSaturationAnalysis(
            model_id="m", namespace="ns",
            total_replicas=sum(r[1] for r in rows),
            should_scale_up=should_up, scale_down_safe=down_safe,
            variant_analyses=analyses,
        )
        out = SaturationAnalyzerV1().calculate_saturation_targets(
            analysis, states
        )
        return out, rows

    # STABLE rows: desired is unset (0) or equals current, and the
    # metrics replica count matches current — the ±1 rule applies only
    # in this branch (a transitioning model re-asserts desired instead)
    stable_rows = st.lists(
        st.tuples(
            st.floats(min_value=1.0, max_value=100.0),  # cost
            st.integers(min_value=1, max_value=10),     # current
            st.sampled_from([0, -1]),                   # desired marker
            st.just(0),                                 # pending
        ),
        min_size=1, max_size=5,
    ).map(lambda rows: [
        (c, cur, cur if d == -1 else 0, p) for c, cur, d, p in rows
    ])

    @given(rows=stable_rows, up=st.booleans(), down=st.booleans())
    @settings(max_examples=200, deadline=None)
    def test_stable_model_moves_at_most_one_replica(self, rows, up, down):
        out, rows = self._targets(rows, should_up=up, down_safe=down)
        deltas = [out[f"v{i}"] - cur for i, (_, cur, _, _) in enumerate(rows)]
        assert all(abs(d) <= 1 for d in deltas)
        assert sum(1 for d in deltas if d != 0) <= 1
        assert all(out[f"v{i}"] >= 0 for i in range(len(rows)))

    @given(rows=stable_rows)
    @settings(max_examples=200, deadline=None)
    def test_transition_freezes_other_variants(self, rows):
        """One transitioning variant blocks NEW decisions model-wide:
        its own target re-asserts the in-flight desired, everyone else
        holds current (analyzer.go:314-371)."""
        cost, cur, _, p = rows[0]
        rows = [(cost, cur, cur + 1, p)] + rows[1:]
        out, rows = self._targets(rows, should_up=True, down_safe=True)
        assert out["v0"] == cur + 1  # keeps driving toward desired
        for i, (_, cur_i, _, _) in enumerate(rows[1:], start=1):
            assert out[f"v{i}"] == cur_i  # no new decision while moving

    @given(costs=st.lists(
        st.floats(min_value=1.0, max_value=100.0),
        min_size=2, max_size=5, unique=True,
    ))
    @settings(max_examples=150, deadline=None)
    def test_scale_up_goes_to_cheapest(self, costs):
        rows = [(c, 2, 2, 0) for c in costs]  # steady (desired==current)
        out, rows = self._targets(rows, should_up=True)
        cheapest = costs.index(min(costs))
        for i in range(len(costs)):
            expected = 3 if i == cheapest else 2
            assert out[f"v{i}"] == expected

    @given(costs=st.lists(
        st.floats(min_value=1.0, max_value=100.0),
        min_size=2, max_size=5, unique=True,
    ))
    @settings(max_examples=150, deadline=None)
    def test_scale_down_takes_most_expensive(self, costs):
        rows = [(c, 2, 2, 0) for c in costs]
        out, rows = self._targets(rows, down_safe=True)
        priciest = costs.index(max(costs))
        for i in range(len(costs)):
            expected = 1 if i == priciest else 2
            assert out[f"v{i}"] == expected
