"""vLLM replica simulator — the llm-d-inference-sim analog.

The reference tests against ghcr.io/llm-d/llm-d-inference-sim (a vLLM
emulator exposing real vllm:* metric names). This module is the
in-process equivalent: a discrete-time queueing simulation of one vLLM
replica whose service behavior follows the same ITL model the Inferno
analyzer assumes (iteration time = α + β·batch — reference
docs/design/modeling-optimization.md:52), with KV-token accounting matching
the V2 capacity model (k1 = blocks × block_size tokens on 288 GB HBM3E).

ServiceProfile defaults are MI355X-shaped and are REPLACED by measured
values from wva_amd.calibration when run on hardware.
"""
from __future__ import annotations

import random
from dataclasses import dataclass
from typing import Deque, Dict, List, Optional, Sequence, Tuple
from collections import deque


@dataclass
class ServiceProfile:
    """Decode/prefill service parameters for one replica.

    alpha/beta follow ITL(batch) = alpha + beta·batch (ms). Defaults are
    placeholders in the emulator; wva_amd/calibration measures real values
    on MI355X (profiles/ carries the measured curves).
    """

    alpha_ms: float = 12.0
    beta_ms: float = 0.35
    max_num_seqs: int = 256
    num_gpu_blocks: int = 120_000  # MI355X 288 GB-scale Llama-3-8B default
    block_size: int = 16
    prefill_tokens_per_s: float = 50_000.0
    prefix_cache_hit_rate: float = 0.0
    # optional measured per-batch ITL table: [(batch, itl_ms), ...] sorted
    # by batch. When set, itl_ms() interpolates the table instead of the
    # linear α+β·B model — required for MoE decode, whose ITL is concave
    # (all experts active by B≈8; see docs/calibration.md, Mixtral section).
    itl_table: Optional[Tuple[Tuple[int, float], ...]] = None

    @property
    def kv_capacity_tokens(self) -> int:
        return self.num_gpu_blocks * self.block_size

    def itl_ms(self, batch: int) -> float:
        batch = max(batch, 1)
        if self.itl_table:
            return self._interp_itl(batch)
        return self.alpha_ms + self.beta_ms * batch

    def _interp_itl(self, batch: int) -> float:
        table = self.itl_table
        if batch <= table[0][0]:
            return table[0][1]
        for (b0, t0), (b1, t1) in zip(table, table[1:]):
            if batch <= b1:
                frac = (batch - b0) / (b1 - b0)
                return t0 + frac * (t1 - t0)
        # beyond the measured range: extend with the last segment's slope
        (b0, t0), (b1, t1) = table[-2], table[-1]
        slope = (t1 - t0) / (b1 - b0)
        return t1 + slope * (batch - b1)

    @classmethod
    def from_itl_table(
        cls,
        batches: Sequence[int],
        itl_ms: Sequence[float],
        **kwargs,
    ) -> "ServiceProfile":
        """Profile from a measured per-batch ITL curve.

        `batches`/`itl_ms` come straight from a calibration JSON
        (profiles/calibration_*.json: keys `batch_sizes` / `itl_ms`).
        α/β are also derived by least squares over the table so consumers
        that only understand the linear model (the V2 analyzer's
        ITLAtBatch and Inferno's ServiceParmsSpec) get the best linear
        approximation, while the simulator itself interpolates exactly.
        """
        if len(batches) != len(itl_ms) or len(batches) < 2:
            raise ValueError("need >=2 (batch, itl) pairs of equal length")
        pairs = tuple(sorted(zip((int(b) for b in batches), itl_ms)))
        n = len(pairs)
        sx = sum(b for b, _ in pairs)
        sy = sum(t for _, t in pairs)
        sxx = sum(b * b for b, _ in pairs)
        sxy = sum(b * t for b, t in pairs)
        denom = n * sxx - sx * sx
        beta = (n * sxy - sx * sy) / denom if denom else 0.0
        alpha = (sy - beta * sx) / n
        return cls(
            alpha_ms=max(alpha, 0.0),
            beta_ms=max(beta, 0.0),
            itl_table=pairs,
            **kwargs,
        )


@dataclass
class RequestSpec:
    input_tokens: int = 100
    output_tokens: int = 50
    arrival_time: float = 0.0


@dataclass
class _RunningRequest:
    spec: RequestSpec
    generated: int = 0
    prefill_remaining: float = 0.0

    def kv_tokens(self) -> int:
        return self.spec.input_tokens + self.generated


@dataclass
class CompletedRequest:
    spec: RequestSpec
    start_time: float = 0.0
    first_token_time: float = 0.0
    finish_time: float = 0.0

    @property
    def ttft(self) -> float:
        return self.first_token_time - self.spec.arrival_time

    @property
    def itl(self) -> float:
        if self.spec.output_tokens <= 1:
            return 0.0
        return (self.finish_time - self.first_token_time) / (
            self.spec.output_tokens - 1
        )


class ReplicaSim:
    """One vLLM replica: waiting queue + running batch + KV accounting."""

    def __init__(self, pod_name: str, profile: ServiceProfile):
        self.pod_name = pod_name
        self.profile = profile
        self.waiting: Deque[_RunningRequest] = deque()
        self.running: List[_RunningRequest] = []
        self._start_times: Dict[int, float] = {}
        # cumulative counters (vllm:* counter semantics)
        self.request_success_total = 0
        self.prompt_tokens_sum = 0
        self.prompt_tokens_count = 0
        self.generation_tokens_sum = 0
        self.generation_tokens_count = 0
        self.prefix_cache_hits = 0
        self.prefix_cache_queries = 0
        self.ttft_sum = 0.0
        self.ttft_count = 0
        self.tpot_sum = 0.0
        self.tpot_count = 0
        self.completed: List[CompletedRequest] = []
        # peaks since last scrape (max_over_time[1m] approximation)
        self._peak_kv = 0.0
        self._peak_queue = 0

    # --- metrics ---

    def kv_tokens_in_use(self) -> int:
        return sum(r.kv_tokens() for r in self.running)

    def kv_cache_usage(self) -> float:
        cap = self.profile.kv_capacity_tokens
        return min(self.kv_tokens_in_use() / cap, 1.0) if cap else 0.0

    def num_requests_waiting(self) -> int:
        return len(self.waiting)

    def num_requests_running(self) -> int:
        return len(self.running)

    def observe_peaks(self) -> None:
        self._peak_kv = max(self._peak_kv, self.kv_cache_usage())
        self._peak_queue = max(self._peak_queue, len(self.waiting))

    def peak_kv_and_reset(self) -> float:
        self.observe_peaks()
        v = self._peak_kv
        self._peak_kv = self.kv_cache_usage()
        return v

    def peak_queue_and_reset(self) -> int:
        self.observe_peaks()
        v = self._peak_queue
        self._peak_queue = len(self.waiting)
        return v

    # --- load ---

    def submit(self, spec: RequestSpec) -> None:
        self.waiting.append(_RunningRequest(spec=spec))

    def load(self) -> float:
        """Dispatch weight: running + waiting (least-loaded LB)."""
        return len(self.running) + len(self.waiting)

    # --- simulation ---

    def _try_admit(self, now: float) -> None:
        while self.waiting and len(self.running) < self.profile.max_num_seqs:
            candidate = self.waiting[0]
            # a preemption victim re-acquires prompt + generated-so-far
            projected = self.kv_tokens_in_use() + candidate.kv_tokens()
            # vLLM-style watermark: keep a little headroom for decode growth
            if projected > 0.98 * self.profile.kv_capacity_tokens:
                break
            self.waiting.popleft()
            if candidate.generated == 0:
                # prefix cache: a hit fraction of prompt tokens skips prefill
                self.prefix_cache_queries += 1
                hit = random.random() < self.profile.prefix_cache_hit_rate
                if hit:
                    self.prefix_cache_hits += 1
                candidate.prefill_remaining = (
                    candidate.spec.input_tokens * (0.0 if hit else 1.0)
                )
            # else: re-admission after recompute preemption — keep the
            # prefill_remaining the eviction assigned (prompt + generated;
            # recomputed tokens never hit the prefix cache)
            self._start_times.setdefault(id(candidate), now)
            self.running.append(candidate)

    def _preempt_if_over_capacity(self) -> None:
        """vLLM recompute-style preemption: when decode growth pushes
        the running set past the KV pool, the NEWEST request is evicted
        back to the head of the waiting queue and must re-prefill its
        prompt + generated-so-far on re-admission (its KV blocks free
        immediately). Keeps kv_tokens_in_use ≤ capacity like the real
        engine — the admission watermark alone only bounds the INPUT
        footprint."""
        cap = self.profile.kv_capacity_tokens
        while len(self.running) > 1 and self.kv_tokens_in_use() > cap:
            victim = self.running.pop()
            victim.prefill_remaining = float(victim.kv_tokens())
            self.waiting.appendleft(victim)

    def step(self, now: float, dt: float) -> List[CompletedRequest]:
        """Advance the replica by dt seconds of simulated time."""
        self._try_admit(now)
        self._preempt_if_over_capacity()
        finished: List[CompletedRequest] = []
        if self.running:
            # prefill first (chunked-prefill approximation: prefill shares
            # the iteration budget, modeled as a separate token bucket)
            prefill_budget = self.profile.prefill_tokens_per_s * dt
            for r in self.running:
                if r.prefill_remaining > 0 and prefill_budget > 0:
                    used = min(r.prefill_remaining, prefill_budget)
                    r.prefill_remaining -= used
                    prefill_budget -= used

            decoding = [r for r in self.running if r.prefill_remaining <= 0]
            if decoding:
                batch = len(decoding)
                iter_time_s = self.profile.itl_ms(batch) / 1000.0
                iterations = dt / iter_time_s if iter_time_s > 0 else 0
                whole = int(iterations)
                frac = iterations - whole
                if random.random() < frac:
                    whole += 1
                for r in decoding:
                    if whole <= 0:
                        break
                    first = r.generated == 0
                    r.generated = min(
                        r.generated + whole, r.spec.output_tokens
                    )
                    if first and r.generated > 0:
                        self.ttft_sum += now + iter_time_s - r.spec.arrival_time
                        self.ttft_count += 1
                        r._first_token_time = now + iter_time_s  # type: ignore[attr-defined]
                    if r.generated >= r.spec.output_tokens:
                        finished.append(self._complete(r, now + dt))
        self.running = [
            r for r in self.running if r.generated < r.spec.output_tokens
        ]
        self._try_admit(now + dt)
        # decode growth during this step may have crossed the pool —
        # preempt before the state is observed (scrapes see ≤ capacity,
        # exactly like the real engine's per-iteration scheduler)
        self._preempt_if_over_capacity()
        self.observe_peaks()
        return finished

    def _complete(self, r: _RunningRequest, finish_time: float) -> CompletedRequest:
        self.request_success_total += 1
        self.prompt_tokens_sum += r.spec.input_tokens
        self.prompt_tokens_count += 1
        self.generation_tokens_sum += r.spec.output_tokens
        self.generation_tokens_count += 1
        first_token_time = getattr(r, "_first_token_time", finish_time)
        comp = CompletedRequest(
            spec=r.spec,
            start_time=self._start_times.pop(id(r), r.spec.arrival_time),
            first_token_time=first_token_time,
            finish_time=finish_time,
        )
        if r.spec.output_tokens > 1:
            self.tpot_sum += comp.itl * (r.spec.output_tokens - 1)
            self.tpot_count += r.spec.output_tokens - 1
        self.completed.append(comp)
        return comp
