"""Property tests for the vLLM replica simulator.

The emulator is the substrate of the headline bench (docs/benchmark.md)
and of every e2e test, so its bookkeeping must hold under ANY workload,
not just the hand-picked ones in test_emulator.py. hypothesis drives
random request mixes / step sizes and checks the structural invariants:

* conservation: every submitted request is in exactly one of
  waiting / running / completed;
* counters are cumulative (vllm:* counter semantics — never decrease);
* token accounting: completed prompt/generation sums match the specs;
* KV pool: while more than one request runs, usage stays within the
  pool (the recompute-preemption contract; a SOLE runner may exceed it
  — the livelock guard, see ReplicaSim._preempt_if_over_capacity);
* latency sanity: TTFT ≥ 0 and ITL > 0 for every completion.
"""
import random

from hypothesis import given, settings, strategies as st

from wva_amd.emulator.vllm_sim import ReplicaSim, RequestSpec, ServiceProfile


request_specs = st.lists(
    st.tuples(
        st.integers(min_value=1, max_value=400),    # input_tokens
        st.integers(min_value=1, max_value=120),    # output_tokens
        st.floats(min_value=0.0, max_value=5.0),    # arrival_time
    ),
    min_size=1,
    max_size=25,
)

profiles = st.tuples(
    st.floats(min_value=1.0, max_value=30.0),       # alpha_ms
    st.floats(min_value=0.01, max_value=2.0),       # beta_ms
    st.integers(min_value=1, max_value=16),         # max_num_seqs
    st.integers(min_value=30, max_value=400),       # num_gpu_blocks
)


def _run(specs, prof_tuple, dt, seconds=30.0, seed=0):
    random.seed(seed)
    alpha, beta, seqs, blocks = prof_tuple
    sim = ReplicaSim("p0", ServiceProfile(
        alpha_ms=alpha, beta_ms=beta, max_num_seqs=seqs,
        num_gpu_blocks=blocks, prefill_tokens_per_s=1e6,
    ))
    pending = sorted(
        (RequestSpec(input_tokens=i, output_tokens=o, arrival_time=a)
         for i, o, a in specs),
        key=lambda s: s.arrival_time,
    )
    t, submitted = 0.0, 0
    prev_success = 0
    while t < seconds:
        while pending and pending[0].arrival_time <= t:
            sim.submit(pending.pop(0))
            submitted += 1
        sim.step(t, dt)

        # conservation: nothing is lost or duplicated
        in_flight = len(sim.waiting) + len(sim.running)
        assert in_flight + len(sim.completed) == submitted

        # counters only grow
        assert sim.request_success_total >= prev_success
        prev_success = sim.request_success_total
        assert sim.request_success_total == len(sim.completed)

        # KV pool respected whenever preemption is allowed to act
        if len(sim.running) > 1:
            assert sim.kv_tokens_in_use() <= sim.profile.kv_capacity_tokens
        assert 0.0 <= sim.kv_cache_usage() <= 1.0

        t += dt
    return sim, submitted


class TestReplicaSimInvariants:
    @settings(max_examples=60, deadline=None)
    @given(specs=request_specs, prof=profiles,
           dt=st.sampled_from([0.02, 0.05, 0.25]))
    def test_structural_invariants_hold(self, specs, prof, dt):
        sim, submitted = _run(specs, prof, dt)
        # whatever completed has exact token accounting
        assert sim.prompt_tokens_sum == sum(
            c.spec.input_tokens for c in sim.completed
        )
        assert sim.generation_tokens_sum == sum(
            c.spec.output_tokens for c in sim.completed
        )
        assert sim.prompt_tokens_count == len(sim.completed)

    @settings(max_examples=60, deadline=None)
    @given(specs=request_specs, prof=profiles)
    def test_completion_latencies_sane(self, specs, prof):
        sim, _ = _run(specs, prof, dt=0.05)
        for c in sim.completed:
            assert c.ttft >= 0.0
            assert c.finish_time >= c.first_token_time
            if c.spec.output_tokens > 1:
                assert c.itl > 0.0

    @settings(max_examples=40, deadline=None)
    @given(specs=request_specs, prof=profiles)
    def test_everything_fitting_eventually_completes(self, specs, prof):
        # requests whose peak KV exceeds the pool can stall legally; a
        # workload of individually-fitting requests must fully drain
        alpha, beta, seqs, blocks = prof
        cap = blocks * 16
        specs = [
            (i, o, a) for i, o, a in specs if i + o <= 0.9 * cap
        ] or [(1, 1, 0.0)]
        sim, submitted = _run(specs, prof, dt=0.05, seconds=600.0)
        assert len(sim.completed) == submitted

    @settings(max_examples=30, deadline=None)
    @given(specs=request_specs, prof=profiles,
           seed=st.integers(min_value=0, max_value=2**16))
    def test_deterministic_given_seed(self, specs, prof, seed):
        a, _ = _run(specs, prof, dt=0.05, seed=seed)
        b, _ = _run(specs, prof, dt=0.05, seed=seed)
        assert a.request_success_total == b.request_success_total
        assert a.ttft_sum == b.ttft_sum
        assert a.tpot_sum == b.tpot_sum
        assert [c.finish_time for c in a.completed] == \
            [c.finish_time for c in b.completed]
