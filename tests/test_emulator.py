"""Emulator unit tests: ReplicaSim queueing dynamics, ClusterSim
deployment controller, EPP admission bound, event recording.
"""
import pytest

from wva_amd.api.types import ObjectMeta
from wva_amd.api.types import (
    CrossVersionObjectReference,
    VariantAutoscaling,
    VariantAutoscalingSpec,
)
from wva_amd.emulator.cluster_sim import ClusterSim
from wva_amd.emulator.vllm_sim import ReplicaSim, RequestSpec, ServiceProfile
from wva_amd.kube.fake import FakeCluster
from wva_amd.kube.objects import Container, Deployment, PodTemplateSpec


def replica(alpha=10.0, beta=0.5, seqs=8, blocks=100):
    return ReplicaSim("p0", ServiceProfile(
        alpha_ms=alpha, beta_ms=beta, max_num_seqs=seqs,
        num_gpu_blocks=blocks, prefill_tokens_per_s=1e9,
    ))


class TestReplicaSim:
    def test_request_completes(self):
        r = replica()
        r.submit(RequestSpec(input_tokens=10, output_tokens=5, arrival_time=0))
        t = 0.0
        done = []
        while not done and t < 5.0:
            done = r.step(t, 0.05)
            t += 0.05
        assert len(done) == 1
        assert r.request_success_total == 1
        assert r.generation_tokens_sum == 5
        assert r.prompt_tokens_sum == 10
        comp = done[0]
        assert comp.ttft >= 0
        assert comp.itl > 0

    def test_kv_accounting(self):
        r = replica(blocks=100)  # 1600 token slots
        r.submit(RequestSpec(input_tokens=100, output_tokens=50, arrival_time=0))
        r.step(0, 0.001)
        assert r.num_requests_running() == 1
        assert r.kv_tokens_in_use() >= 100
        assert 0 < r.kv_cache_usage() < 1

    def test_admission_respects_batch_cap(self):
        r = replica(seqs=2)
        for i in range(5):
            r.submit(RequestSpec(arrival_time=0))
        r.step(0, 0.001)
        assert r.num_requests_running() == 2
        assert r.num_requests_waiting() == 3

    def test_admission_respects_kv_capacity(self):
        r = replica(seqs=100, blocks=20)  # 320 token slots
        for i in range(5):
            r.submit(RequestSpec(input_tokens=100, output_tokens=10,
                                 arrival_time=0))
        r.step(0, 0.001)
        # 3 × 100 tokens would exceed the 0.98 watermark of 320
        assert r.num_requests_running() <= 3

    def test_itl_model(self):
        """Measured completion time follows ITL(batch) = alpha + beta·batch."""
        r = replica(alpha=10.0, beta=1.0, seqs=4, blocks=10000)
        for i in range(4):
            r.submit(RequestSpec(input_tokens=1, output_tokens=20,
                                 arrival_time=0))
        t = 0.0
        while r.request_success_total < 4 and t < 10:
            r.step(t, 0.01)
            t += 0.01
        # ITL at batch 4 = 14 ms; 20 tokens ≈ 0.28 s
        assert 0.2 < t < 0.6

    def test_peak_tracking(self):
        r = replica(seqs=1)
        for i in range(4):
            r.submit(RequestSpec(arrival_time=0))
        r.step(0, 0.001)
        assert r.peak_queue_and_reset() >= 3


class TestClusterSimController:
    def _mk(self, replicas=2, delay=10.0):
        c = FakeCluster()
        d = Deployment(
            metadata=ObjectMeta(name="v", namespace="ns"),
            replicas=replicas,
            selector={"app": "v"},
            template=PodTemplateSpec(labels={"app": "v"},
                                     containers=[Container()]),
        )
        c.create(d)
        sim = ClusterSim(c, pod_ready_delay_s=delay)
        sim.register_variant("m", "ns", "v", ServiceProfile())
        return c, sim

    def test_pods_created_and_become_ready(self):
        c, sim = self._mk(replicas=2, delay=10.0)
        sim.reconcile_deployments()
        d = c.get("Deployment", "ns", "v")
        assert d.status.replicas == 2
        assert d.status.ready_replicas == 0
        sim.advance(11.0)
        sim.reconcile_deployments()  # readiness updates on the next pass
        d = c.get("Deployment", "ns", "v")
        assert d.status.ready_replicas == 2
        assert len(c.list("Pod", namespace="ns")) == 2

    def test_scale_down_deletes_pods(self):
        c, sim = self._mk(replicas=3, delay=0.0)
        sim.reconcile_deployments()
        c.scale("Deployment", "ns", "v", 1)
        sim.reconcile_deployments()
        assert len(c.list("Pod", namespace="ns")) == 1

    def test_scheduler_queue_drains_on_ready(self):
        c, sim = self._mk(replicas=1, delay=5.0)
        sim.reconcile_deployments()
        model = sim.model("m", "ns")
        sim.submit_request(model, RequestSpec(arrival_time=0))
        assert len(model.scheduler_queue) == 1
        sim.advance(6.0)
        sim.advance(0.1)
        assert len(model.scheduler_queue) == 0


class TestEvents:
    def test_target_not_found_records_warning(self):
        from prometheus_client import CollectorRegistry

        from wva_amd.app import build_app
        from wva_amd.config.config import Config

        c = FakeCluster()
        c.create(VariantAutoscaling(
            metadata=ObjectMeta(name="va1", namespace="ns"),
            spec=VariantAutoscalingSpec(
                scale_target_ref=CrossVersionObjectReference(name="missing"),
                model_id="m",
            ),
        ))
        cfg = Config()
        cfg.mark_bootstrap_complete()
        app = build_app(c, cfg, source=_NullSource(),
                        metrics_registry=CollectorRegistry(),
                        start_engines=False)
        app.va_reconciler.reconcile("ns", "va1")
        va = c.get("VariantAutoscaling", "ns", "va1")
        from wva_amd.api import conditions as cond

        assert cond.is_condition_false(va, "TargetResolved")
        assert any(e.reason == "TargetNotFound" for e in c.events)


class _NullSource:
    def name(self):
        return "prometheus"

    def query_list(self):
        from wva_amd.collector.query_template import QueryList

        return QueryList()

    def refresh(self, spec):
        return {}

    def get(self, query, params):
        return None


class TestITLTableProfile:
    """Measured per-batch ITL tables (MoE concave curves) — see
    docs/calibration.md Mixtral section."""

    MIXTRAL = ([1, 8, 32, 64], [23.86, 42.56, 47.15, 48.58])

    def test_interpolation_exact_at_knots(self):
        p = ServiceProfile.from_itl_table(*self.MIXTRAL)
        for b, t in zip(*self.MIXTRAL):
            assert p.itl_ms(b) == pytest.approx(t)

    def test_interpolation_between_knots(self):
        p = ServiceProfile.from_itl_table(*self.MIXTRAL)
        mid = p.itl_ms(20)  # between 8 and 32
        assert 42.56 < mid < 47.15

    def test_extrapolation_uses_last_slope(self):
        p = ServiceProfile.from_itl_table(*self.MIXTRAL)
        slope = (48.58 - 47.15) / (64 - 32)
        assert p.itl_ms(128) == pytest.approx(48.58 + slope * 64)

    def test_below_first_knot_clamps(self):
        p = ServiceProfile.from_itl_table(*self.MIXTRAL)
        assert p.itl_ms(0) == pytest.approx(23.86)

    def test_linear_parms_derived(self):
        p = ServiceProfile.from_itl_table(*self.MIXTRAL)
        assert p.alpha_ms > 0
        assert p.beta_ms > 0
        # linear model must at least be within the table's range at B=32
        assert 20 < p.alpha_ms + p.beta_ms * 32 < 60

    def test_without_table_linear(self):
        p = ServiceProfile(alpha_ms=10.0, beta_ms=0.5)
        assert p.itl_ms(20) == pytest.approx(20.0)

    def test_rejects_short_table(self):
        with pytest.raises(ValueError):
            ServiceProfile.from_itl_table([1], [10.0])

    def test_unsorted_input_sorted(self):
        p = ServiceProfile.from_itl_table([32, 1, 8], [47.15, 23.86, 42.56])
        assert p.itl_ms(1) == pytest.approx(23.86)
        assert p.itl_ms(32) == pytest.approx(47.15)


class TestMoEDensePath:
    def test_dense_matches_sparse(self):
        import torch
        from wva_amd.calibration.moe_model import TINY_MOE, MixtralDecodeModel

        m = MixtralDecodeModel(TINY_MOE, max_batch=8, max_seq=32, device="cpu")
        h2 = torch.randn(8, TINY_MOE.hidden_size, dtype=torch.bfloat16)
        layer = m.layers[0]
        logits = h2.float() @ layer.w_router.t().float()
        w, sel = torch.topk(logits, TINY_MOE.top_k, dim=-1)
        w = torch.softmax(w, dim=-1).to(h2.dtype)
        dense = m._moe_mlp_dense(layer, h2, w, sel)
        sparse = m._moe_mlp_sparse(layer, h2, w, sel)
        torch.testing.assert_close(
            dense.float(), sparse.float(), atol=3e-2, rtol=3e-2
        )


class TestEmulatorPhysics:
    """The bench substrate must obey queueing physics — these pin the
    emulator against first principles, independent of any analyzer."""

    def _stack(self, replicas=2, **prof_kwargs):
        from wva_amd.emulator.cluster_sim import ClusterSim
        from wva_amd.emulator.vllm_sim import ServiceProfile
        from wva_amd.kube.fake import FakeCluster
        from wva_amd.kube.objects import (
            Container, Deployment, PodTemplateSpec,
        )
        from wva_amd.api.types import ObjectMeta

        cluster = FakeCluster()
        cluster.create(Deployment(
            metadata=ObjectMeta(name="v", namespace="ns"),
            replicas=replicas,
            selector={"app": "v"},
            template=PodTemplateSpec(
                labels={"app": "v"},
                containers=[Container(requests={"amd.com/gpu": "1"})],
            ),
        ))
        prof = ServiceProfile(**{
            "alpha_ms": 10.0, "beta_ms": 1.0, "max_num_seqs": 16,
            "num_gpu_blocks": 50_000, **prof_kwargs,
        })
        sim = ClusterSim(cluster, warm_start=True, seed=3)
        sim.register_variant("m", "ns", "v", prof)
        sim.reconcile_deployments()
        return sim, sim.model("m", "ns"), prof

    def _run(self, sim, model, qps, seconds, dt=0.25):
        from wva_amd.emulator.workload import constant_qps

        prof = constant_qps(qps)
        for _ in range(int(seconds / dt)):
            sim.generate_arrivals(model, prof, dt, 100, 50)
            sim.advance(dt)

    def test_steady_state_throughput_matches_offered(self):
        """Under-loaded system: completions/s ≈ offered qps (flow
        conservation)."""
        sim, model, prof = self._stack(replicas=2)
        self._run(sim, model, qps=5.0, seconds=60)
        n0, t0 = len(model.completed), sim.now
        self._run(sim, model, qps=5.0, seconds=120)
        rate = (len(model.completed) - n0) / (sim.now - t0)
        assert rate == pytest.approx(5.0, rel=0.15)

    def test_overload_throughput_capped_at_service_rate(self):
        """Overloaded replicas complete at their saturated service rate
        (B/ITL(B)/O per replica), regardless of offered load."""
        sim, model, prof = self._stack(replicas=1)
        cap = prof.max_num_seqs / (
            prof.itl_ms(prof.max_num_seqs) / 1000.0
        ) / 50.0
        self._run(sim, model, qps=cap * 4, seconds=30)
        n0, t0 = len(model.completed), sim.now
        self._run(sim, model, qps=cap * 4, seconds=60)
        rate = (len(model.completed) - n0) / (sim.now - t0)
        assert rate == pytest.approx(cap, rel=0.2)
        assert rate < cap * 1.2

    def test_ttft_grows_under_backlog(self):
        sim, model, prof = self._stack(replicas=1)
        self._run(sim, model, qps=1.0, seconds=30)
        light = [c.ttft for c in model.completed[-10:]]
        cap = prof.max_num_seqs / (
            prof.itl_ms(prof.max_num_seqs) / 1000.0
        ) / 50.0
        self._run(sim, model, qps=cap * 3, seconds=60)
        heavy = [c.ttft for c in model.completed[-10:]]
        assert sum(heavy) / len(heavy) > 5 * (sum(light) / len(light))

    def test_itl_reflects_batch_occupancy(self):
        """Measured per-request ITL ≈ the profile's ITL at the running
        occupancy (the service curve actually drives the sim)."""
        sim, model, prof = self._stack(replicas=1, max_num_seqs=8)
        self._run(sim, model, qps=1.0, seconds=90)
        recents = model.completed[-15:]
        itls = [c.itl * 1000 for c in recents if c.spec.output_tokens > 1]
        mean_itl = sum(itls) / len(itls)
        # light load: occupancy between 1 and ~4 → ITL within the curve
        assert prof.itl_ms(1) * 0.9 <= mean_itl <= prof.itl_ms(8) * 1.1

    def test_kv_capacity_bounds_admission(self):
        """Tiny KV pool: concurrent tokens never exceed capacity (the
        k1 constraint the V2 analyzer scales on)."""
        sim, model, prof = self._stack(
            replicas=1, num_gpu_blocks=100, max_num_seqs=64
        )  # 1600-token pool vs ~150-token requests
        cap = prof.kv_capacity_tokens
        peak = 0
        from wva_amd.emulator.workload import constant_qps

        w = constant_qps(50.0)
        for _ in range(200):
            sim.generate_arrivals(model, w, 0.25, 100, 50)
            sim.advance(0.25)
            for _pod, (rep, _r, _d, _n) in list(sim.replicas.items()):
                peak = max(peak, sum(r.kv_tokens() for r in rep.running))
        assert 0 < peak <= cap


class TestRecomputePreemption:
    """vLLM recompute-style preemption: decode growth past the KV pool
    evicts the newest request, which re-prefills prompt + generated on
    re-admission (ReplicaSim._preempt_if_over_capacity)."""

    def _overcommitted(self):
        # 320 token slots; two requests admitted at input=140 each (280
        # < 0.98·320) then decode growth pushes past capacity
        r = replica(seqs=8, blocks=20)
        # 320 slots: each request peaks at 140+60=200 (fits alone) but the
        # pair's decode growth crosses 320, forcing preemption
        for i in range(2):
            r.submit(RequestSpec(
                input_tokens=140, output_tokens=60, arrival_time=0.0
            ))
        return r

    def test_kv_never_exceeds_capacity(self):
        r = self._overcommitted()
        t = 0.0
        for _ in range(200):
            r.step(t, 0.05)
            assert r.kv_tokens_in_use() <= r.profile.kv_capacity_tokens
            t += 0.05

    def test_victim_requeued_and_re_prefills_generated(self):
        r = self._overcommitted()
        t = 0.0
        preempted = None
        for _ in range(200):
            r.step(t, 0.05)
            t += 0.05
            if r.waiting and r.waiting[0].generated > 0:
                preempted = r.waiting[0]
                break
        assert preempted is not None, "over-committed pair never preempted"
        # eviction charges re-prefill for prompt + generated-so-far
        assert preempted.prefill_remaining == pytest.approx(
            float(preempted.spec.input_tokens + preempted.generated)
        )

    def test_readmission_keeps_recompute_prefill(self):
        """_try_admit must not overwrite the eviction-assigned
        prefill_remaining with prompt-only (or prefix-cache-skipped)
        prefill, and must project admission on kv_tokens()."""
        r = replica(seqs=8, blocks=100)  # 1600 slots, plenty of room
        r.profile.prefix_cache_hit_rate = 1.0  # would zero a fresh prefill
        from wva_amd.emulator.vllm_sim import _RunningRequest

        victim = _RunningRequest(
            spec=RequestSpec(input_tokens=100, output_tokens=50), generated=30
        )
        victim.prefill_remaining = 130.0  # as _preempt_if_over_capacity sets
        r.waiting.append(victim)
        queries_before = r.prefix_cache_queries
        r._try_admit(now=1.0)
        assert victim in r.running
        assert victim.prefill_remaining == 130.0
        # recomputed tokens never hit the prefix cache
        assert r.prefix_cache_queries == queries_before

    def test_victim_completes_and_ttft_counted_once(self):
        r = self._overcommitted()
        t = 0.0
        done = []
        for _ in range(4000):
            done += r.step(t, 0.05)
            t += 0.05
            if len(done) == 2:
                break
        assert len(done) == 2
        assert r.ttft_count == 2  # no double TTFT on re-admission

    def test_sole_runner_never_preempted(self):
        # len(running) > 1 guard: a single request over capacity keeps
        # running (evicting it would livelock)
        r = replica(seqs=8, blocks=8)  # 128 slots < 100+100 peak
        r.submit(RequestSpec(input_tokens=100, output_tokens=100,
                             arrival_time=0.0))
        t = 0.0
        done = []
        for _ in range(400):
            done = r.step(t, 0.05)
            if done:
                break
            assert r.num_requests_running() == 1
            assert r.num_requests_waiting() == 0
            t += 0.05
        assert len(done) == 1  # ran to completion despite exceeding pool
