"""Property tests for the remaining numerical/structural primitives:
RFC 7386 merge patch (the #731 status-patch path), the measured-ITL
interpolation profile, and the 3-parameter service-surface fit.
"""
import math

from hypothesis import assume, given, settings, strategies as st

from wva_amd.kube.openapi import merge_patch


# --- RFC 7386 merge patch ---

JSON = st.recursive(
    st.none() | st.booleans() | st.integers(-1000, 1000)
    | st.text(max_size=8),
    lambda children: st.lists(children, max_size=3)
    | st.dictionaries(st.text(max_size=6), children, max_size=3),
    max_leaves=12,
)


def spec_merge(target, patch):
    """Literal transcription of RFC 7386 §2 pseudocode."""
    if isinstance(patch, dict):
        if not isinstance(target, dict):
            target = {}
        else:
            target = dict(target)
        for name, value in patch.items():
            if value is None:
                target.pop(name, None)
            else:
                target[name] = spec_merge(target.get(name), value)
        return target
    return patch


class TestMergePatchRFC7386:
    @settings(max_examples=300, deadline=None)
    @given(target=JSON, patch=JSON)
    def test_matches_spec_pseudocode(self, target, patch):
        assert merge_patch(target, patch) == spec_merge(target, patch)

    @settings(max_examples=200, deadline=None)
    @given(target=JSON, patch=JSON)
    def test_idempotent(self, target, patch):
        once = merge_patch(target, patch)
        assert merge_patch(once, patch) == once

    @settings(max_examples=200, deadline=None)
    @given(target=JSON, patch=JSON)
    def test_target_not_mutated(self, target, patch):
        import copy

        snapshot = copy.deepcopy(target)
        merge_patch(target, patch)
        assert target == snapshot

    @settings(max_examples=200, deadline=None)
    @given(
        target=st.dictionaries(st.text(max_size=6), JSON, max_size=4),
        patch=st.dictionaries(
            st.text(max_size=6),
            st.none() | st.integers(-10, 10),
            max_size=4,
        ),
    )
    def test_null_deletes_everything_it_names(self, target, patch):
        out = merge_patch(target, patch)
        for k, v in patch.items():
            if v is None:
                assert k not in out
            else:
                assert out[k] == v


# --- measured ITL-table interpolation ---

ITL_TABLE = st.lists(
    st.tuples(
        st.integers(min_value=1, max_value=512),
        st.floats(min_value=0.1, max_value=500.0),
    ),
    min_size=2,
    max_size=8,
    unique_by=lambda p: p[0],
)


class TestITLTableInterpolation:
    @settings(max_examples=200, deadline=None)
    @given(table=ITL_TABLE)
    def test_passes_through_measured_points(self, table):
        from wva_amd.emulator.vllm_sim import ServiceProfile

        prof = ServiceProfile.from_itl_table(
            [b for b, _ in table], [t for _, t in table]
        )
        for b, t in table:
            assert math.isclose(prof.itl_ms(b), t, rel_tol=1e-12)

    @settings(max_examples=200, deadline=None)
    @given(table=ITL_TABLE)
    def test_permutation_invariant(self, table):
        from wva_amd.emulator.vllm_sim import ServiceProfile

        a = ServiceProfile.from_itl_table(
            [b for b, _ in table], [t for _, t in table]
        )
        rev = list(reversed(table))
        b_ = ServiceProfile.from_itl_table(
            [b for b, _ in rev], [t for _, t in rev]
        )
        assert a.itl_table == b_.itl_table
        for q in (1, 3, 17, 100, 600):
            assert a.itl_ms(q) == b_.itl_ms(q)

    @settings(max_examples=200, deadline=None)
    @given(table=ITL_TABLE, q=st.integers(min_value=1, max_value=512))
    def test_within_range_bounded_by_neighbors(self, table, q):
        from wva_amd.emulator.vllm_sim import ServiceProfile

        table = sorted(table)
        assume(table[0][0] <= q <= table[-1][0])
        prof = ServiceProfile.from_itl_table(
            [b for b, _ in table], [t for _, t in table]
        )
        v = prof.itl_ms(q)
        for (b0, t0), (b1, t1) in zip(table, table[1:]):
            if b0 <= q <= b1:
                assert min(t0, t1) - 1e-9 <= v <= max(t0, t1) + 1e-9
                break

    @settings(max_examples=100, deadline=None)
    @given(table=ITL_TABLE)
    def test_derived_linear_parms_nonnegative(self, table):
        from wva_amd.emulator.vllm_sim import ServiceProfile

        prof = ServiceProfile.from_itl_table(
            [b for b, _ in table], [t for _, t in table]
        )
        assert prof.alpha_ms >= 0.0
        assert prof.beta_ms >= 0.0


# --- 3-parameter service-surface fit ---

class TestSurfaceFitProperties:
    grids = st.lists(
        st.tuples(
            st.integers(min_value=1, max_value=256),      # batch
            st.integers(min_value=128, max_value=65536),  # ctx
        ),
        min_size=4,
        max_size=12,
        unique=True,
    )

    @settings(max_examples=100, deadline=None)
    @given(
        grid=grids,
        alpha=st.floats(min_value=0.5, max_value=50.0),
        beta=st.floats(min_value=1e-3, max_value=1.0),
        gamma=st.floats(min_value=0.0, max_value=1e-3),
    )
    def test_recovers_exact_planar_data(self, grid, alpha, beta, gamma):
        from wva_amd.calibration.itl_benchmark import fit_itl_surface

        # need rank-3 design: at least two distinct batches and two
        # distinct n·ctx products
        assume(len({b for b, _ in grid}) >= 2)
        assume(len({b * c for b, c in grid}) >= 3)
        pts = [(b, c, alpha + b * (beta + gamma * c)) for b, c in grid]
        a, be, ga, r2 = fit_itl_surface(pts)
        if r2 > 0.999999:  # well-conditioned draw
            assert math.isclose(a, alpha, rel_tol=1e-3, abs_tol=1e-2)
            assert math.isclose(be, beta, rel_tol=1e-3, abs_tol=1e-3)

    @settings(max_examples=100, deadline=None)
    @given(
        grid=grids,
        noise=st.lists(
            st.floats(min_value=-5.0, max_value=5.0), min_size=12, max_size=12
        ),
    )
    def test_outputs_always_physical(self, grid, noise):
        from wva_amd.calibration.itl_benchmark import fit_itl_surface

        pts = [
            (b, c, max(0.1, 10.0 + noise[i % len(noise)]))
            for i, (b, c) in enumerate(grid)
        ]
        a, be, ga, r2 = fit_itl_surface(pts)
        assert be >= 0.0 and ga >= 0.0
        assert math.isfinite(a) and math.isfinite(r2)


# --- CostAwareOptimizer decision invariants ---

from wva_amd.analyzers.interfaces import (
    ACTION_SCALE_DOWN,
    ACTION_SCALE_UP,
    AnalyzerResult,
    VariantCapacity,
    VariantReplicaState,
)
from wva_amd.pipeline.limiter import ModelScalingRequest
from wva_amd.pipeline.optimizer import ACTION_NO_CHANGE, CostAwareOptimizer

variants = st.lists(
    st.tuples(
        st.floats(min_value=0.5, max_value=100.0),       # cost
        st.integers(min_value=0, max_value=20),          # current replicas
        st.one_of(st.just(0.0),
                  st.floats(min_value=100.0, max_value=1e6)),  # per-replica cap
    ),
    min_size=1,
    max_size=5,
)


def _request(vars_, required=0.0, spare=0.0):
    vcs, states = [], []
    for i, (cost, cur, prc) in enumerate(vars_):
        name = f"v{i}"
        vcs.append(VariantCapacity(
            variant_name=name, cost=cost, replica_count=cur,
            per_replica_capacity=prc, total_capacity=prc * cur,
        ))
        states.append(VariantReplicaState(
            variant_name=name, current_replicas=cur,
        ))
    result = AnalyzerResult(
        model_id="m", namespace="ns", variant_capacities=vcs,
        required_capacity=required, spare_capacity=spare,
    )
    return ModelScalingRequest(
        model_id="m", namespace="ns", result=result, variant_states=states,
    )


class TestCostAwareOptimizerProperties:
    @settings(max_examples=200, deadline=None)
    @given(vars_=variants,
           required=st.floats(min_value=1.0, max_value=1e6))
    def test_scale_up_covers_required_capacity(self, vars_, required):
        req = _request(vars_, required=required)
        decisions = CostAwareOptimizer().optimize([req])
        by_name = {d.variant_name: d for d in decisions}
        added = 0.0
        for vc in req.result.variant_capacities:
            d = by_name[vc.variant_name]
            assert d.target_replicas >= 0
            added += (
                (d.target_replicas - d.current_replicas)
                * vc.per_replica_capacity
            )
        if any(prc > 0 for _, _, prc in vars_):
            assert added >= required - 1e-6
        else:
            assert added == 0.0  # nothing can absorb the demand

    @settings(max_examples=200, deadline=None)
    @given(vars_=variants,
           spare=st.floats(min_value=1.0, max_value=1e6))
    def test_scale_down_never_removes_more_than_spare(self, vars_, spare):
        req = _request(vars_, spare=spare)
        decisions = CostAwareOptimizer().optimize([req])
        by_name = {d.variant_name: d for d in decisions}
        removed = 0.0
        for vc in req.result.variant_capacities:
            d = by_name[vc.variant_name]
            assert d.target_replicas >= 0
            assert d.target_replicas <= d.current_replicas  # down only
            removed += (
                (d.current_replicas - d.target_replicas)
                * vc.per_replica_capacity
            )
        assert removed <= spare + 1e-6

    @settings(max_examples=200, deadline=None)
    @given(vars_=variants)
    def test_steady_state_no_changes(self, vars_):
        req = _request(vars_)  # required == spare == 0
        for d in CostAwareOptimizer().optimize([req]):
            assert d.action == ACTION_NO_CHANGE
            assert d.target_replicas == d.current_replicas

    @settings(max_examples=200, deadline=None)
    @given(vars_=variants,
           spare=st.floats(min_value=1e5, max_value=1e9))
    def test_huge_spare_keeps_cheapest_alive(self, vars_, spare):
        """Even unbounded spare capacity never drains the model to zero
        replicas if it had any: the cheapest variant keeps one."""
        req = _request(vars_, spare=spare)
        had_replicas = any(cur > 0 for _, cur, _ in vars_)
        decisions = CostAwareOptimizer().optimize([req])
        if had_replicas:
            assert sum(d.target_replicas for d in decisions) >= 1

    @settings(max_examples=200, deadline=None)
    @given(vars_=variants,
           required=st.floats(min_value=1.0, max_value=1e6),
           spare=st.floats(min_value=1.0, max_value=1e6))
    def test_action_labels_match_targets(self, vars_, required, spare):
        # exercise both branches (required wins when both > 0)
        req = _request(vars_, required=required, spare=spare)
        for d in CostAwareOptimizer().optimize([req]):
            if d.target_replicas > d.current_replicas:
                assert d.action == ACTION_SCALE_UP
            elif d.target_replicas < d.current_replicas:
                assert d.action == ACTION_SCALE_DOWN
            else:
                assert d.action == ACTION_NO_CHANGE

    @settings(max_examples=100, deadline=None)
    @given(
        costs=st.lists(
            st.floats(min_value=1.0, max_value=100.0),
            min_size=2, max_size=5, unique=True,
        ),
        required=st.floats(min_value=1.0, max_value=1e5),
    )
    def test_equal_capacity_prefers_cheapest(self, costs, required):
        """With identical per-replica capacity everywhere, cost-efficiency
        ordering degenerates to cost ordering: the FIRST replicas added go
        to the cheapest variant."""
        prc = 1e6  # one replica more than covers any generated demand
        vars_ = [(c, 1, prc) for c in costs]
        req = _request(vars_, required=required)
        decisions = CostAwareOptimizer().optimize([req])
        cheapest = f"v{costs.index(min(costs))}"
        for d in decisions:
            if d.variant_name == cheapest:
                assert d.target_replicas > d.current_replicas
            else:
                assert d.target_replicas == d.current_replicas


# --- GPU allocation (TypeAllocator + GreedyBySaturation) invariants ---

from wva_amd.analyzers.interfaces import VariantDecision
from wva_amd.pipeline.greedy_saturation import GreedyBySaturation
from wva_amd.pipeline.inventory import TypeAllocator

alloc_decisions = st.lists(
    st.tuples(
        st.integers(min_value=0, max_value=8),    # current replicas
        st.integers(min_value=0, max_value=12),   # extra replicas wanted
        st.integers(min_value=1, max_value=8),    # gpus per replica
        st.floats(min_value=-1e5, max_value=1e5),  # spare capacity
        st.floats(min_value=1.0, max_value=100.0),  # cost
    ),
    min_size=1,
    max_size=6,
)


def _decisions(rows, acc="MI355X"):
    out = []
    for i, (cur, extra, gpr, spare, cost) in enumerate(rows):
        out.append(VariantDecision(
            variant_name=f"v{i}", namespace="ns", model_id="m",
            accelerator_name=acc, cost=cost,
            current_replicas=cur, target_replicas=cur + extra,
            gpus_per_replica=gpr, spare_capacity=spare,
        ))
    return out


class TestAllocationProperties:
    @settings(max_examples=200, deadline=None)
    @given(rows=alloc_decisions, pool=st.integers(min_value=0, max_value=64))
    def test_pool_conserved_and_whole_replicas(self, rows, pool):
        decisions = _decisions(rows)
        allocator = TypeAllocator({"MI355X": pool})
        GreedyBySaturation().allocate(decisions, allocator)
        total_allocated = 0
        for i, d in enumerate(decisions):
            cur, extra, gpr, _, _ = rows[i]
            # whole-replica granularity, no partial-replica leak
            assert d.gpus_allocated % gpr == 0
            assert d.gpus_allocated == (d.target_replicas - cur) * gpr
            # never scales beyond the ask, never below current
            assert cur <= d.target_replicas <= cur + extra
            if d.target_replicas < cur + extra and extra > 0:
                assert d.was_limited
            total_allocated += d.gpus_allocated
        assert total_allocated <= pool
        assert allocator.remaining() == pool - total_allocated
        assert allocator.remaining_for_type("MI355X") >= 0

    @settings(max_examples=200, deadline=None)
    @given(rows=alloc_decisions, pool=st.integers(min_value=1, max_value=64))
    def test_most_saturated_candidate_served_first(self, rows, pool):
        decisions = _decisions(rows)
        allocator = TypeAllocator({"MI355X": pool})
        GreedyBySaturation().allocate(decisions, allocator)
        # order candidates as the algorithm does; once one is limited,
        # every later one gets nothing more than pool leftovers allow —
        # in particular a FULLY limited candidate (0 allocated despite
        # wanting some) means everyone later with >= its gpus/replica
        # got 0 too
        cands = sorted(
            (d for i, d in enumerate(decisions)
             if d.target_replicas + (0) >= 0 and rows[i][1] > 0),
            key=lambda d: (d.spare_capacity, d.cost),
        )
        seen_starved_gpr = None
        for d in cands:
            if seen_starved_gpr is not None \
                    and d.gpus_per_replica >= seen_starved_gpr:
                assert d.gpus_allocated == 0
            if d.gpus_allocated == 0:
                if seen_starved_gpr is None \
                        or d.gpus_per_replica < seen_starved_gpr:
                    seen_starved_gpr = d.gpus_per_replica

    @settings(max_examples=100, deadline=None)
    @given(rows=alloc_decisions,
           hive=st.integers(min_value=1, max_value=8),
           pool=st.integers(min_value=8, max_value=64))
    def test_hive_constraint_rejects_oversized_replicas(self, rows, pool,
                                                        hive):
        decisions = _decisions(rows)
        allocator = TypeAllocator({"MI355X": pool}, {"MI355X": hive})
        GreedyBySaturation().allocate(decisions, allocator)
        for i, d in enumerate(decisions):
            cur, extra, gpr, _, _ = rows[i]
            if gpr > hive:
                assert d.gpus_allocated == 0
                assert d.target_replicas == cur

    @settings(max_examples=100, deadline=None)
    @given(rows=alloc_decisions, pool=st.integers(min_value=0, max_value=64))
    def test_wrong_type_pool_untouched(self, rows, pool):
        decisions = _decisions(rows, acc="MI300X")
        allocator = TypeAllocator({"MI355X": pool})
        GreedyBySaturation().allocate(decisions, allocator)
        assert allocator.remaining_for_type("MI355X") == pool
        for d in decisions:
            assert d.gpus_allocated == 0


# --- V2 token analyzer invariants ---

from wva_amd.analyzers.capacity_store import CapacityKnowledgeStore
from wva_amd.analyzers.interfaces import AnalyzerInput, ReplicaMetrics
from wva_amd.analyzers.saturation_v2 import SaturationAnalyzerV2
from wva_amd.config.saturation import SaturationScalingConfig

replica_metrics = st.lists(
    st.tuples(
        st.integers(min_value=0, max_value=2),           # variant index
        st.integers(min_value=0, max_value=3_000_000),   # kv capacity tokens
        st.floats(min_value=0.0, max_value=1.0),         # kv usage frac
        st.integers(min_value=0, max_value=50),          # queue length
        st.floats(min_value=0.0, max_value=4096.0),      # avg input tokens
        st.floats(min_value=0.0, max_value=1024.0),      # avg output tokens
    ),
    min_size=0,
    max_size=8,
)

v2_config = st.tuples(
    st.floats(min_value=0.05, max_value=1.0),  # scale_up_threshold
    st.floats(min_value=0.05, max_value=1.0),  # scale_down_boundary
    st.floats(min_value=0.1, max_value=1.0),   # kv_cache_threshold
)


def _v2_input(metrics_rows, cfg_row):
    up, down, kv = cfg_row
    cfg = SaturationScalingConfig(analyzer_name="saturation-v2")
    cfg.apply_defaults()
    cfg.scale_up_threshold = max(up, down)  # validated invariant: up > down
    cfg.scale_down_boundary = min(up, down) * 0.99
    cfg.kv_cache_threshold = kv
    rms, states = [], {}
    from wva_amd.analyzers.interfaces import VariantReplicaState

    for i, (vi, cap, usage, qlen, avg_in, avg_out) in enumerate(metrics_rows):
        name = f"var{vi}"
        rms.append(ReplicaMetrics(
            pod_name=f"p{i}", variant_name=name, namespace="ns",
            model_id="m", accelerator_name="MI355X",
            kv_cache_usage=usage, queue_length=qlen,
            total_kv_capacity_tokens=cap,
            tokens_in_use=int(usage * cap),
            avg_input_tokens=avg_in, avg_output_tokens=avg_out,
        ))
        s = states.setdefault(name, VariantReplicaState(variant_name=name))
        s.current_replicas += 1
    return AnalyzerInput(
        model_id="m", namespace="ns", replica_metrics=rms,
        variant_states=list(states.values()), config=cfg,
    )


class TestV2AnalyzerProperties:
    @settings(max_examples=200, deadline=None)
    @given(rows=replica_metrics, cfg=v2_config)
    def test_result_always_finite_and_consistent(self, rows, cfg):
        inp = _v2_input(rows, cfg)
        res = SaturationAnalyzerV2(CapacityKnowledgeStore()).analyze(inp)
        assert math.isfinite(res.total_supply) and res.total_supply >= 0
        assert math.isfinite(res.total_demand) and res.total_demand >= 0
        assert math.isfinite(res.utilization) and res.utilization >= 0
        assert res.required_capacity >= 0.0
        assert res.spare_capacity >= 0.0
        # supply equals the sum over variants
        assert math.isclose(
            res.total_supply,
            sum(vc.total_capacity for vc in res.variant_capacities),
            rel_tol=1e-9, abs_tol=1e-6,
        )
        # utilization is exactly demand/supply (or 0 on empty supply)
        if res.total_supply > 0:
            assert math.isclose(
                res.utilization, res.total_demand / res.total_supply,
                rel_tol=1e-9,
            )
        else:
            assert res.utilization == 0.0

    @settings(max_examples=200, deadline=None)
    @given(rows=replica_metrics, cfg=v2_config)
    def test_never_up_and_down_at_once(self, rows, cfg):
        """required > 0 and spare > 0 simultaneously would make the
        optimizer direction ambiguous; the up>down config invariant
        forbids it whenever there are no pending replicas (anticipated
        supply == current supply)."""
        inp = _v2_input(rows, cfg)
        res = SaturationAnalyzerV2(CapacityKnowledgeStore()).analyze(inp)
        if res.required_capacity > 1e-9:
            assert res.spare_capacity <= 1e-6 * max(res.total_supply, 1.0)

    @settings(max_examples=100, deadline=None)
    @given(rows=replica_metrics, cfg=v2_config)
    def test_per_replica_capacity_bounded_by_k1(self, rows, cfg):
        """Effective capacity = min(k1, k2) ≤ k1 = kv_threshold × total."""
        inp = _v2_input(rows, cfg)
        res = SaturationAnalyzerV2(CapacityKnowledgeStore()).analyze(inp)
        caps_by_variant = {}
        for vi, cap, *_ in rows:
            caps_by_variant.setdefault(f"var{vi}", []).append(cap)
        kv = inp.config.kv_cache_threshold
        for vc in res.variant_capacities:
            caps = caps_by_variant.get(vc.variant_name)
            if caps and vc.per_replica_capacity > 0:
                assert vc.per_replica_capacity <= max(caps) * kv + 1.0

    @settings(max_examples=100, deadline=None)
    @given(cfg=v2_config)
    def test_empty_input_is_calm(self, cfg):
        """No replicas → no demand, no supply, no scaling pressure."""
        inp = _v2_input([], cfg)
        res = SaturationAnalyzerV2(CapacityKnowledgeStore()).analyze(inp)
        assert res.total_supply == 0.0
        assert res.total_demand == 0.0
        assert res.required_capacity == 0.0
        assert res.spare_capacity == 0.0


# --- Enforcer fail-safe invariants ---

from wva_amd.analyzers.interfaces import VariantSaturationAnalysis
from wva_amd.config.scale_to_zero import ModelScaleToZeroConfig
from wva_amd.pipeline.enforcer import Enforcer

targets_maps = st.dictionaries(
    st.sampled_from(["v-a", "v-b", "v-c"]),
    st.integers(min_value=0, max_value=10),
    min_size=1,
    max_size=3,
)


def _s2z(enabled):
    return {"default": ModelScaleToZeroConfig(
        enable_scale_to_zero=enabled, retention_period="10m",
    )}


class TestEnforcerProperties:
    @settings(max_examples=150, deadline=None)
    @given(targets=targets_maps,
           count=st.floats(min_value=0.0, max_value=1e6))
    def test_scale_to_zero_only_on_exact_zero_traffic(self, targets, count):
        enf = Enforcer(lambda m, ns, r: count)
        out, applied = enf.enforce_policy(
            "m", "ns", dict(targets), [], _s2z(True),
        )
        if count > 0:
            assert out == targets
            assert applied is False
        else:
            assert all(v == 0 for v in out.values())
            assert applied is True

    @settings(max_examples=150, deadline=None)
    @given(targets=targets_maps)
    def test_query_error_never_zeroes(self, targets):
        def boom(m, ns, r):
            raise RuntimeError("prometheus down")

        out, applied = Enforcer(boom).enforce_policy(
            "m", "ns", dict(targets), [], _s2z(True),
        )
        assert out == targets  # fail safe: keep whatever the optimizer said
        assert applied is False

    @settings(max_examples=150, deadline=None)
    @given(targets=targets_maps,
           costs=st.lists(st.floats(min_value=1.0, max_value=100.0),
                          min_size=3, max_size=3))
    def test_minimum_replica_floor_when_disabled(self, targets, costs):
        analyses = [
            VariantSaturationAnalysis(variant_name=n, cost=c)
            for n, c in zip(["v-a", "v-b", "v-c"], costs)
        ]
        out, applied = Enforcer(lambda m, ns, r: 0.0).enforce_policy(
            "m", "ns", dict(targets), analyses, _s2z(False),
        )
        assert sum(out.values()) >= 1  # never zero with s2z disabled
        if sum(targets.values()) > 0:
            assert out == targets and applied is False
        else:
            assert applied is True
            # the floor landed on the cheapest variant in the target map
            kept = [n for n, v in out.items() if v > 0]
            assert len(kept) == 1
            cost_of = {a.variant_name: a.cost for a in analyses}
            best = min(
                targets, key=lambda n: (cost_of.get(n, 10.0), n)
            )
            assert kept[0] == best


# --- DedupWorkQueue semantics ---

class TestDedupWorkQueueProperties:
    @settings(max_examples=150, deadline=None)
    @given(events=st.lists(
        st.tuples(st.sampled_from(["a", "b", "c", "d"]),
                  st.sampled_from(["ns1", "ns2"])),
        min_size=1, max_size=30,
    ))
    def test_dedups_while_pending_and_preserves_first_seen_order(
        self, events
    ):
        from wva_amd.runtime.manager import DedupWorkQueue

        q = DedupWorkQueue()
        fn = lambda ns, name: None  # noqa: E731
        for name, ns in events:
            q.put((fn, ns, name))
        drained = []
        while True:
            item = q.get(timeout=0.0)
            if item is None:
                break
            drained.append((item[2], item[1]))
        # exactly the distinct items, in first-appearance order
        seen, expected = set(), []
        for e in events:
            if e not in seen:
                seen.add(e)
                expected.append(e)
        assert drained == expected
        assert q.empty()

    def test_readd_after_pickup_requeues(self):
        from wva_amd.runtime.manager import DedupWorkQueue

        q = DedupWorkQueue()
        fn = lambda ns, name: None  # noqa: E731
        q.put((fn, "ns", "x"))
        item = q.get(timeout=0.0)
        assert item is not None
        q.put((fn, "ns", "x"))  # in-flight item may be queued again
        assert q.get(timeout=0.0) == item


# --- vLLM args parsing + PromQL templating robustness ---

from wva_amd.analyzers.deployment_parser import (
    parse_vllm_args,
    split_shell_string,
)
from wva_amd.collector.query_template import QueryTemplate, escape_promql_value
from wva_amd.kube.objects import Container, Deployment, PodTemplateSpec


class TestVLLMArgsParsing:
    def test_multiline_block_scalar_command(self):
        """`sh -c` with a YAML `|` block scalar: newlines and `\\`
        continuations separate args like a real shell."""
        cmd = (
            "vllm serve meta-llama/Llama-3.1-8B \\\n"
            "  --max-num-seqs 512 \\\n"
            "\t--block-size 16\n"
            "  --kv-cache-dtype fp8\r\n"
        )
        d = Deployment(
            metadata=__import__(
                "wva_amd.api.types", fromlist=["ObjectMeta"]
            ).ObjectMeta(name="d", namespace="ns"),
            template=PodTemplateSpec(containers=[
                Container(name="vllm", command=["sh", "-c", cmd]),
            ]),
        )
        p = parse_vllm_args(d)
        assert p.max_num_seqs == 512
        assert p.block_size == 16
        assert p.kv_cache_dtype == "fp8"

    @settings(max_examples=200, deadline=None)
    @given(s=st.text(max_size=120))
    def test_splitter_never_crashes_and_never_emits_empty(self, s):
        toks = split_shell_string(s)
        assert all(isinstance(t, str) and t for t in toks)
        assert all(t != "\\" for t in toks)

    @settings(max_examples=150, deadline=None)
    @given(args=st.lists(st.text(max_size=40), max_size=12))
    def test_arbitrary_container_args_never_crash(self, args):
        from wva_amd.api.types import ObjectMeta

        d = Deployment(
            metadata=ObjectMeta(name="d", namespace="ns"),
            template=PodTemplateSpec(containers=[
                Container(name="vllm", args=list(args)),
            ]),
        )
        p = parse_vllm_args(d)
        assert p.block_size >= 1
        assert p.max_num_seqs >= 1
        assert p.tensor_parallel_size >= 1


class TestPromQLTemplateSafety:
    @settings(max_examples=300, deadline=None)
    @given(value=st.text(max_size=60))
    def test_escaped_value_cannot_break_out_of_label_matcher(self, value):
        """Whatever a model/namespace name contains, the rendered query
        keeps it inside the quoted label value: the escaped form has no
        unescaped double quote."""
        esc = escape_promql_value(value)
        i, n = 0, len(esc)
        while i < n:
            if esc[i] == "\\":
                i += 2  # escape consumes the next char
                continue
            assert esc[i] != '"'
            i += 1

    @settings(max_examples=200, deadline=None)
    @given(value=st.text(max_size=60))
    def test_render_roundtrip_preserves_value(self, value):
        """Unescaping the rendered label value gives back the input —
        escaping is injective (no two models collide onto one query)."""
        t = QueryTemplate(
            name="q", template='metric{model_name="{{.model}}"}',
            params=["model"],
        )
        out = t.render({"model": value})
        prefix, suffix = 'metric{model_name="', '"}'
        assert out.startswith(prefix) and out.endswith(suffix)
        inner = out[len(prefix):-len(suffix)]
        unescaped = inner.replace('\\"', '"').replace("\\\\", "\\")
        # precise unescape (left-to-right)
        res, i = [], 0
        while i < len(inner):
            if inner[i] == "\\" and i + 1 < len(inner):
                res.append(inner[i + 1])
                i += 2
            else:
                res.append(inner[i])
                i += 1
        assert "".join(res) == value


# --- Prometheus exposition label escaping roundtrip ---

from wva_amd.collector.pod_scraping_source import parse_prometheus_text


def _escape_label_value(v: str) -> str:
    """Exposition-format escape, per the Prometheus text format spec."""
    return v.replace("\\", "\\\\").replace('"', '\\"').replace("\n", "\\n")


class TestExpositionLabelRoundtrip:
    @settings(max_examples=300, deadline=None)
    @given(value=st.text(
        alphabet=st.characters(blacklist_categories=("Cs",)), max_size=40,
    ))
    def test_escaped_label_value_roundtrips(self, value):
        line = f'vllm:metric{{model="{_escape_label_value(value)}"}} 1.0\n'
        vals = parse_prometheus_text(line)
        assert len(vals) == 1
        assert vals[0].labels["model"] == value

    def test_escaped_backslash_before_n_is_not_newline(self):
        # raw bytes: model="a\\n" → escaped backslash + literal n
        vals = parse_prometheus_text('m{model="a\\\\n"} 1\n')
        assert vals[0].labels["model"] == "a\\n"
        # while a real \n escape IS a newline
        vals = parse_prometheus_text('m{model="a\\nb"} 1\n')
        assert vals[0].labels["model"] == "a\nb"

    @settings(max_examples=150, deadline=None)
    @given(text=st.text(max_size=200))
    def test_garbage_never_crashes(self, text):
        for mv in parse_prometheus_text(text):
            assert math.isfinite(mv.value) or math.isnan(mv.value) or \
                math.isinf(mv.value)
            assert "__name__" in mv.labels


# --- OpenAPI validator spec conformance ---

from wva_amd.kube.openapi import validate


class TestOpenAPIValidator:
    def test_pattern_is_partial_match_per_json_schema(self):
        schema = {"type": "string", "pattern": "abc"}
        assert validate(schema, "xxabcxx") == []     # partial match OK
        assert validate(schema, "xyz") != []
        anchored = {"type": "string", "pattern": "^abc$"}
        assert validate(anchored, "abc") == []
        assert validate(anchored, "xxabcxx") != []

    @settings(max_examples=200, deadline=None)
    @given(obj=JSON)
    def test_untyped_schema_accepts_anything(self, obj):
        assert validate({}, obj) == []

    @settings(max_examples=200, deadline=None)
    @given(
        n=st.integers(min_value=-100, max_value=100),
        lo=st.integers(min_value=-50, max_value=50),
        hi=st.integers(min_value=-50, max_value=50),
    )
    def test_integer_bounds_exact(self, n, lo, hi):
        schema = {"type": "integer", "minimum": lo, "maximum": hi}
        ok = validate(schema, n) == []
        assert ok == (lo <= n <= hi)

    @settings(max_examples=150, deadline=None)
    @given(obj=JSON)
    def test_bool_is_not_an_integer(self, obj):
        # kube treats booleans and integers as distinct
        errs = validate({"type": "integer"}, True)
        assert errs
        errs = validate({"type": "number"}, False)
        assert errs

    @settings(max_examples=150, deadline=None)
    @given(
        required=st.lists(st.sampled_from(["a", "b", "c"]),
                          max_size=3, unique=True),
        present=st.dictionaries(
            st.sampled_from(["a", "b", "c"]),
            st.integers(-5, 5) | st.none(),
            max_size=3,
        ),
    )
    def test_required_means_present_and_non_null(self, required, present):
        schema = {"type": "object", "required": list(required)}
        errs = validate(schema, present)
        expect_missing = [
            r for r in required
            if r not in present or present[r] is None
        ]
        assert len(errs) == len(expect_missing)
