"""V2 token-based analyzer + capacity store tests.

Mirrors reference saturation_v2/{analyzer,capacity_store}_test.go coverage,
with MI355X-scale (288 GB HBM3E) block counts exercised explicitly.
"""
from wva_amd.analyzers.capacity_store import CapacityKnowledgeStore, CapacityRecord
from wva_amd.analyzers.deployment_parser import VLLMEngineParams
from wva_amd.analyzers.interfaces import (
    AnalyzerInput,
    ReplicaMetrics,
    SchedulerQueueMetrics,
    VariantReplicaState,
)
from wva_amd.analyzers.saturation_v2 import (
    SaturationAnalyzerV2,
    estimate_capacity_from_params,
    estimate_scheduler_queue_demand,
)
from wva_amd.config.saturation import SaturationScalingConfig
from wva_amd.kube.objects import Container, Deployment, PodTemplateSpec

# MI355X: 288 GB HBM3E. With ~0.9 utilization, Llama-3-8B bf16 KV ≈ 128 KiB
# per token → ~1.9M KV token slots per GPU; num_gpu_blocks ≈ 120k at
# block_size 16. Use that scale in tests.
MI355X_BLOCKS = 120_000
BLOCK = 16
MI355X_KV_TOKENS = MI355X_BLOCKS * BLOCK  # 1.92M tokens


def v2cfg():
    return SaturationScalingConfig.from_dict({"analyzerName": "saturation"})


def rmet(
    pod="p0",
    variant="v1",
    kv=0.5,
    q=0,
    blocks=MI355X_BLOCKS,
    avg_in=100.0,
    avg_out=50.0,
    accel="MI355X",
    cost=10.0,
    hit=0.0,
):
    total = blocks * BLOCK
    return ReplicaMetrics(
        pod_name=pod,
        variant_name=variant,
        kv_cache_usage=kv,
        queue_length=q,
        accelerator_name=accel,
        cost=cost,
        num_gpu_blocks=blocks,
        block_size=BLOCK,
        total_kv_capacity_tokens=total,
        tokens_in_use=int(kv * total),
        avg_input_tokens=avg_in,
        avg_output_tokens=avg_out,
        prefix_cache_hit_rate=hit,
    )


def vstate(name="v1", current=1, pending=0, desired=0, gpus=1):
    return VariantReplicaState(
        variant_name=name,
        current_replicas=current,
        pending_replicas=pending,
        desired_replicas=desired,
        gpus_per_replica=gpus,
    )


class TestK2Derivation:
    def test_formula(self):
        p = VLLMEngineParams(
            effective_max_batched_tokens=8192, max_num_seqs=256
        )
        # N_steady = min(8192*50/150, 256) = min(2730.67, 256) = 256
        # k2 = 256 * (100 + 25) = 32000
        assert estimate_capacity_from_params(p, 100, 50) == 32000

    def test_batch_bound(self):
        p = VLLMEngineParams(
            effective_max_batched_tokens=1024, max_num_seqs=256
        )
        # N_steady = min(1024*50/150, 256) = 341.33→341.33>256? 341>256 → 256?
        # 1024*50/150 = 341.33 > 256 → N=256... use smaller B
        p2 = VLLMEngineParams(effective_max_batched_tokens=300, max_num_seqs=256)
        # N = min(300*50/150=100, 256)=100 → k2 = 100*(125)=12500
        assert estimate_capacity_from_params(p2, 100, 50) == 12500

    def test_zero_output(self):
        p = VLLMEngineParams(effective_max_batched_tokens=8192)
        assert estimate_capacity_from_params(p, 100, 0) == 0

    def test_none_params(self):
        assert estimate_capacity_from_params(None, 100, 50) == 0


class TestAnalyze:
    def test_skips_replicas_without_capacity_data(self):
        an = SaturationAnalyzerV2()
        m = rmet()
        m.total_kv_capacity_tokens = 0
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=[m],
                variant_states=[vstate()],
                config=v2cfg(),
            )
        )
        assert res.total_supply == 0

    def test_mi355x_scale_k1(self):
        """k1 at MI355X 288 GB scale: 1.92M tokens × 0.80 threshold."""
        an = SaturationAnalyzerV2()
        m = rmet(kv=0.10)
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=[m],
                variant_states=[vstate()],
                config=v2cfg(),
            )
        )
        # no deployment record in store → k2 derivation impossible →
        # k2 = k1 = 1,536,000 tokens (memory-bound at 288 GB scale)
        vc = res.variant_capacities[0]
        assert vc.per_replica_capacity == int(MI355X_KV_TOKENS * 0.80)
        # demand = tokensInUse (kv=0.1 → 192000) + 0 queue
        assert vc.total_demand == int(0.10 * MI355X_KV_TOKENS)

    def test_k2_derived_with_deployment_params(self):
        """With a deployment record in the store, k2 derives from vLLM args:
        N_steady = min(8192·50/150, 256) = 256; k2 = 256·125 = 32000."""
        store = CapacityKnowledgeStore()
        an = SaturationAnalyzerV2(store)
        d = Deployment(
            template=PodTemplateSpec(containers=[Container(args=[])])
        )
        store.load_from_deployment("ns", "m", "v1", "MI355X", 1, d)
        m = rmet(kv=0.10)
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=[m],
                variant_states=[vstate()],
                config=v2cfg(),
            )
        )
        assert res.variant_capacities[0].per_replica_capacity == 32000

    def test_k1_bound_when_no_workload_stats(self):
        an = SaturationAnalyzerV2()
        m = rmet(kv=0.10, avg_in=0.0, avg_out=0.0)
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=[m],
                variant_states=[vstate()],
                config=v2cfg(),
            )
        )
        # no derivation possible → k2 falls back to k1 → effective = k1
        k1 = int(MI355X_KV_TOKENS * 0.80)
        assert res.variant_capacities[0].per_replica_capacity == k1

    def test_observed_k2_at_queue_saturation(self):
        an = SaturationAnalyzerV2()
        m = rmet(kv=0.5, q=10)  # q >= threshold 5, tokensInUse > 0
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=[m],
                variant_states=[vstate()],
                config=v2cfg(),
            )
        )
        tokens_in_use = int(0.5 * MI355X_KV_TOKENS)
        assert res.variant_capacities[0].per_replica_capacity == tokens_in_use
        # history recorded: subsequent non-saturated call uses it
        m2 = rmet(kv=0.2, q=0)
        res2 = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=[m2],
                variant_states=[vstate()],
                config=v2cfg(),
            )
        )
        assert res2.variant_capacities[0].per_replica_capacity == tokens_in_use

    def test_required_capacity_scale_up_signal(self):
        an = SaturationAnalyzerV2()
        # replica at 79% of k1 with queue → high demand, all stats empty →
        # k2=k1=1.536M; demand near capacity
        m = rmet(kv=0.79, q=0, avg_in=0.0, avg_out=0.0)
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=[m],
                variant_states=[vstate()],
                config=v2cfg(),
            )
        )
        k1 = int(MI355X_KV_TOKENS * 0.80)
        demand = int(0.79 * MI355X_KV_TOKENS)
        expected_required = demand / 0.85 - k1
        assert abs(res.required_capacity - expected_required) < 1.0
        assert res.spare_capacity == 0.0

    def test_spare_capacity_scale_down_signal(self):
        an = SaturationAnalyzerV2()
        metrics = [
            rmet(pod=f"p{i}", kv=0.05, avg_in=0.0, avg_out=0.0) for i in range(3)
        ]
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=metrics,
                variant_states=[vstate(current=3)],
                config=v2cfg(),
            )
        )
        k1 = int(MI355X_KV_TOKENS * 0.80)
        demand = 3 * int(0.05 * MI355X_KV_TOKENS)
        assert abs(res.spare_capacity - (3 * k1 - demand / 0.70)) < 1.0
        assert res.required_capacity == 0.0

    def test_pending_replicas_count_into_anticipated_supply(self):
        an = SaturationAnalyzerV2()
        m = rmet(kv=0.79, avg_in=0.0, avg_out=0.0)
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=[m],
                variant_states=[vstate(current=2, pending=1)],
                config=v2cfg(),
            )
        )
        # anticipated supply = (ready 1 + pending 1) × perReplica → doubles
        k1 = int(MI355X_KV_TOKENS * 0.80)
        demand = int(0.79 * MI355X_KV_TOKENS)
        expected_required = max(demand / 0.85 - 2 * k1, 0)
        assert abs(res.required_capacity - expected_required) < 1.0

    def test_median_capacity_across_replicas(self):
        an = SaturationAnalyzerV2()
        # three replicas, all k1-bound with different kv usage (same capacity)
        metrics = [
            rmet(pod="p0", kv=0.1, avg_in=0.0, avg_out=0.0),
            rmet(pod="p1", kv=0.2, avg_in=0.0, avg_out=0.0, blocks=100_000),
            rmet(pod="p2", kv=0.3, avg_in=0.0, avg_out=0.0, blocks=80_000),
        ]
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=metrics,
                variant_states=[vstate(current=3)],
                config=v2cfg(),
            )
        )
        # per-replica = median(k1 values) = k1(100k blocks)
        assert res.variant_capacities[0].per_replica_capacity == int(
            100_000 * BLOCK * 0.80
        )

    def test_scheduler_queue_demand(self):
        metrics = [rmet(avg_in=200, avg_out=100, hit=0.5)]
        sq = SchedulerQueueMetrics(queue_size=10, queue_bytes=4000)
        d = estimate_scheduler_queue_demand(sq, metrics)
        # input = max(4000/4=1000, 10*200=2000) * (1-0.5) = 1000
        # output = 10*100 = 1000
        assert d == 2000.0

    def test_scheduler_queue_none(self):
        assert estimate_scheduler_queue_demand(None, []) == 0.0


class TestZeroReplicaEstimation:
    def test_stored_live_record(self):
        store = CapacityKnowledgeStore()
        an = SaturationAnalyzerV2(store)
        store.update(
            "ns", "m", "v1",
            CapacityRecord(
                accelerator_name="MI355X",
                gpu_count=1,
                effective_capacity=500_000,
                learned_from="live",
            ),
        )
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=[],
                variant_states=[vstate(current=0)],
                config=v2cfg(),
            )
        )
        assert res.variant_capacities[0].per_replica_capacity == 500_000
        assert res.variant_capacities[0].total_capacity == 0  # 0 ready replicas

    def test_compatible_sibling(self):
        store = CapacityKnowledgeStore()
        an = SaturationAnalyzerV2(store)
        params = VLLMEngineParams(effective_max_batched_tokens=8192)
        # the zero-replica variant has a deployment record w/o capacity
        store.update(
            "ns", "m", "new-variant",
            CapacityRecord(
                accelerator_name="MI355X",
                gpu_count=8,
                vllm_params=params,
                learned_from="deployment",
                effective_capacity=0,
            ),
        )
        # a live sibling in ANOTHER namespace with same hw+params
        store.update(
            "other-ns", "m", "old-variant",
            CapacityRecord(
                accelerator_name="MI355X",
                gpu_count=8,
                vllm_params=params,
                learned_from="live",
                effective_capacity=900_000,
            ),
        )
        res = an.analyze(
            AnalyzerInput(
                model_id="m",
                namespace="ns",
                replica_metrics=[],
                variant_states=[vstate(name="new-variant", current=0, gpus=8)],
                config=v2cfg(),
            )
        )
        assert res.variant_capacities[0].per_replica_capacity == 900_000


class TestCapacityStore:
    def test_live_not_overwritten_by_deployment(self):
        store = CapacityKnowledgeStore()
        store.update(
            "ns", "m", "v",
            CapacityRecord(learned_from="live", effective_capacity=42),
        )
        d = Deployment(template=PodTemplateSpec(containers=[Container()]))
        store.load_from_deployment("ns", "m", "v", "MI355X", 1, d)
        assert store.get("ns", "m", "v").learned_from == "live"

    def test_load_from_deployment(self):
        store = CapacityKnowledgeStore()
        d = Deployment(
            template=PodTemplateSpec(
                containers=[
                    Container(args=["--num-gpu-blocks-override=120000",
                                    "--block-size=16"])
                ]
            )
        )
        store.load_from_deployment("ns", "m", "v", "MI355X", 8, d)
        rec = store.get("ns", "m", "v")
        assert rec.learned_from == "deployment"
        assert rec.total_kv_capacity_tokens == 120000 * 16
        assert rec.gpu_count == 8
        assert rec.effective_capacity == 8192  # EffectiveMaxBatchedTokens floor

    def test_find_compatible_prefers_live(self):
        store = CapacityKnowledgeStore()
        params = VLLMEngineParams(effective_max_batched_tokens=8192)
        store.update(
            "a", "m", "v1",
            CapacityRecord(
                accelerator_name="MI355X", gpu_count=1, vllm_params=params,
                learned_from="deployment", effective_capacity=100,
            ),
        )
        store.update(
            "b", "m", "v2",
            CapacityRecord(
                accelerator_name="MI355X", gpu_count=1, vllm_params=params,
                learned_from="live", effective_capacity=200,
            ),
        )
        best = store.find_compatible("m", "MI355X", 1, params)
        assert best.learned_from == "live"

    def test_find_compatible_gpu_count_mismatch(self):
        store = CapacityKnowledgeStore()
        params = VLLMEngineParams()
        store.update(
            "a", "m", "v1",
            CapacityRecord(
                accelerator_name="MI355X", gpu_count=8, vllm_params=params,
                learned_from="live", effective_capacity=100,
            ),
        )
        assert store.find_compatible("m", "MI355X", 1, params) is None

    def test_eviction(self):
        store = CapacityKnowledgeStore()
        store.update("a", "m", "v1", CapacityRecord(effective_capacity=1))
        assert store.evict_stale(-1) == 1
        assert len(store) == 0


class TestK2HistoryChain:
    """Reference analyzer_test.go:88-170: the observed→history chain and
    output-length bucketing."""

    def _analyze_once(self, an, metrics, states=None):
        return an.analyze(AnalyzerInput(
            model_id="m", namespace="ns",
            replica_metrics=metrics,
            variant_states=states or [vstate(current=1)],
            config=v2cfg(),
        ))

    def test_observed_k2_stored_then_reused_after_queue_drops(self):
        an = SaturationAnalyzerV2(CapacityKnowledgeStore())
        # saturated: queue >= threshold(5) → observe k2 = tokens_in_use
        sat = rmet(kv=0.3, q=8)
        self._analyze_once(an, [sat])
        observed = sat.tokens_in_use
        # queue drops below threshold → the HISTORICAL k2 drives capacity
        calm = rmet(kv=0.3, q=0)
        res = self._analyze_once(an, [calm])
        vc = res.variant_capacities[0]
        k1 = int(calm.total_kv_capacity_tokens * 0.80)
        assert vc.per_replica_capacity == float(min(observed, k1))
        assert vc.per_replica_capacity == float(observed)  # k2 < k1 here

    def test_output_length_buckets_are_independent(self):
        an = SaturationAnalyzerV2(CapacityKnowledgeStore())
        # short-output saturation observation (bucket for avg_out=50)
        self._analyze_once(an, [rmet(kv=0.3, q=8, avg_out=50.0)])
        short_keys = set(an._compute_capacity_history)
        # long-output saturation observation lands in a DIFFERENT bucket
        self._analyze_once(an, [rmet(kv=0.35, q=8, avg_out=800.0)])
        long_keys = set(an._compute_capacity_history) - short_keys
        assert short_keys and long_keys
        assert short_keys.isdisjoint(long_keys)

    def test_rolling_average_window(self):
        an = SaturationAnalyzerV2(CapacityKnowledgeStore())
        vals = [0.30, 0.40, 0.50]
        for kv in vals:
            self._analyze_once(an, [rmet(kv=kv, q=8)])
        (key, ra), = an._compute_capacity_history.items()
        expect = sum(
            int(kv * MI355X_KV_TOKENS) for kv in vals
        ) / len(vals)
        assert abs(ra.average() - expect) < 1.0


class TestPendingReplicaAsymmetry:
    """Reference analyzer_test.go:226-263: pending replicas raise the
    ANTICIPATED supply (scale-up damping) but must not inflate the
    spare-capacity (scale-down) side."""

    def test_pending_not_in_scale_down_spare(self):
        an = SaturationAnalyzerV2(CapacityKnowledgeStore())
        # light demand on 2 ready replicas
        metrics = [rmet(pod="p0", kv=0.05, q=0), rmet(pod="p1", kv=0.05, q=0)]
        base = an.analyze(AnalyzerInput(
            model_id="m", namespace="ns", replica_metrics=metrics,
            variant_states=[vstate(current=2, pending=0)], config=v2cfg(),
        ))
        with_pending = an.analyze(AnalyzerInput(
            model_id="m", namespace="ns", replica_metrics=metrics,
            variant_states=[vstate(current=4, pending=2)], config=v2cfg(),
        ))
        # spare (scale-down side) unchanged by pending replicas: supply
        # counts READY replicas only
        assert with_pending.total_supply == base.total_supply
        assert with_pending.spare_capacity == base.spare_capacity
        # but required (scale-up side) sees the anticipated supply
        assert with_pending.required_capacity <= base.required_capacity


class TestSchedulerQueueDemandDetails:
    """Reference analyzer_test.go:538-645."""

    def test_prefix_cache_hits_reduce_input_demand(self):
        m_nohit = [rmet(hit=0.0)]
        m_hit = [rmet(hit=0.6)]
        sq = SchedulerQueueMetrics(queue_size=10, queue_bytes=0)
        d0 = estimate_scheduler_queue_demand(sq, m_nohit)
        d1 = estimate_scheduler_queue_demand(sq, m_hit)
        # input part (10×100) shrinks by 60%; output part (10×50) doesn't
        assert d0 == 10 * 100 + 10 * 50
        assert d1 == 10 * 100 * 0.4 + 10 * 50

    def test_max_of_bytes_and_count_estimates(self):
        metrics = [rmet()]
        # bytes/4 = 2000 > count×avgIn = 1000 → bytes wins
        sq = SchedulerQueueMetrics(queue_size=10, queue_bytes=8000)
        assert estimate_scheduler_queue_demand(sq, metrics) == 2000 + 500
        # bytes/4 = 100 < count×avgIn = 1000 → count wins
        sq = SchedulerQueueMetrics(queue_size=10, queue_bytes=400)
        assert estimate_scheduler_queue_demand(sq, metrics) == 1000 + 500

    def test_steady_state_between_thresholds(self):
        """Reference analyzer_test.go:518-537: utilization between
        scaleDownBoundary (0.70) and scaleUpThreshold (0.85) → neither
        scale-up nor scale-down signals."""
        an = SaturationAnalyzerV2(CapacityKnowledgeStore())
        # choose kv usage so demand/supply ≈ 0.78: supply = k1 = 0.8·total
        # demand = kv·total ⇒ kv = 0.78·0.8 = 0.624
        res = an.analyze(AnalyzerInput(
            model_id="m", namespace="ns",
            replica_metrics=[rmet(kv=0.624, q=0)],
            variant_states=[vstate(current=1)],
            config=v2cfg(),
        ))
        assert 0.70 < res.utilization < 0.85
        assert res.required_capacity == 0.0
        assert res.spare_capacity == 0.0


class TestMedianHelper:
    def test_empty(self):
        from wva_amd.analyzers.saturation_v2 import _median

        assert _median([]) == 0

    def test_odd_even(self):
        from wva_amd.analyzers.saturation_v2 import _median

        assert _median([3, 1, 2]) == 2
        assert _median([4, 1, 3, 2]) == 2  # floor of (2+3)/2
