"""Sim-backed MetricsSource.

Implements the MetricsSource protocol directly over the ClusterSim —
the in-process analog of Prometheus scraping the vLLM emulator pods. Each
registered query name is answered from simulation state with the same
label shape the PromQL queries would return, so the
ReplicaMetricsCollector and engines run unmodified against either source.

max_over_time[1m] semantics are approximated by peak-since-last-scrape
tracking in ReplicaSim; rate[5m] ratios are computed from windowed counter
snapshots kept here.
"""
from __future__ import annotations

import time
from collections import deque
from typing import Deque, Dict, List, Optional, Tuple

from ..collector.query_template import QueryList
from ..collector import registration as reg
from ..collector.source import MetricResult, MetricValue, RefreshSpec
from ..config.scale_to_zero import parse_go_duration
from .cluster_sim import ClusterSim, ModelSim

RATE_WINDOW_S = 300.0


class SimMetricsSource:
    def __init__(self, sim: ClusterSim):
        self.sim = sim
        self._query_list = QueryList()
        # pod → deque[(sim_time, prompt_sum, prompt_cnt, gen_sum, gen_cnt,
        #              hits, queries)]
        self._counter_history: Dict[str, Deque[Tuple]] = {}

    def name(self) -> str:
        return "prometheus"

    def query_list(self) -> QueryList:
        return self._query_list

    # --- helpers ---

    def _model(self, model_id: str, namespace: str) -> Optional[ModelSim]:
        return self.sim.models.get(f"{model_id}|{namespace}")

    def _pods_of_model(self, model: ModelSim):
        for pod_name, (sim, ready_at, dname, ns) in self.sim.replicas.items():
            if ns == model.namespace and dname in model.profiles:
                if self.sim.now >= ready_at:
                    yield pod_name, sim

    def _snapshot_counters(self, pod_name: str, sim) -> None:
        hist = self._counter_history.setdefault(pod_name, deque(maxlen=4096))
        hist.append((
            self.sim.now,
            sim.prompt_tokens_sum, sim.prompt_tokens_count,
            sim.generation_tokens_sum, sim.generation_tokens_count,
            sim.prefix_cache_hits, sim.prefix_cache_queries,
        ))

    def _rate_ratio(
        self, pod_name: str, sum_idx: int, cnt_idx: int
    ) -> Optional[float]:
        hist = self._counter_history.get(pod_name)
        if not hist or len(hist) < 2:
            return None
        newest = hist[-1]
        oldest = None
        for entry in hist:
            if newest[0] - entry[0] <= RATE_WINDOW_S:
                oldest = entry
                break
        if oldest is None or oldest is newest:
            oldest = hist[0]
        d_sum = newest[sum_idx] - oldest[sum_idx]
        d_cnt = newest[cnt_idx] - oldest[cnt_idx]
        if d_cnt <= 0:
            return None
        return d_sum / d_cnt

    # --- MetricsSource ---

    def refresh(self, spec: RefreshSpec) -> Dict[str, MetricResult]:
        out: Dict[str, MetricResult] = {}
        model_id = spec.params.get("modelID", "")
        namespace = spec.params.get("namespace", "")
        for query in spec.queries:
            try:
                out[query] = MetricResult(
                    query=query,
                    values=self._answer(query, model_id, namespace, spec.params),
                    fetched_at=time.time(),
                )
            except Exception as e:  # noqa: BLE001
                out[query] = MetricResult(query=query, error=e)
        return out

    def get(self, query: str, params: Dict[str, str]) -> Optional[MetricResult]:
        res = self.refresh(RefreshSpec(queries=[query], params=params))
        return res.get(query)

    # --- per-query answers ---

    def _answer(
        self, query: str, model_id: str, namespace: str, params: Dict[str, str]
    ) -> List[MetricValue]:
        ts = time.time()
        if query in (reg.QUERY_SCHEDULER_QUEUE_SIZE, reg.QUERY_SCHEDULER_QUEUE_BYTES):
            total = 0
            for model in self.sim.models.values():
                if model.model_id != model_id:
                    continue
                if query == reg.QUERY_SCHEDULER_QUEUE_SIZE:
                    total += len(model.scheduler_queue)
                else:
                    total += sum(4 * s.input_tokens for s in model.scheduler_queue)
            return [MetricValue(value=float(total), timestamp=ts)]

        model = self._model(model_id, namespace)
        if model is None:
            return []

        if query == reg.QUERY_MODEL_ARRIVAL_RATE:
            # true arrivals over the window — what the registered PromQL
            # (completion rate + queue-depth derivative) estimates: a
            # completions-only rate goes stale under backlog (arrivals >
            # completions while under-provisioned) and overshoots during
            # drain, destabilizing rate-based sizing
            window = 120.0
            cutoff = self.sim.now - window
            count = sum(
                1 for c in model.completed
                if c.spec.arrival_time >= cutoff
            )
            count += sum(
                1 for spec in model.scheduler_queue
                if spec.arrival_time >= cutoff
            )
            for _pod, rep in self._pods_of_model(model):
                count += sum(
                    1 for r in rep.waiting
                    if r.spec.arrival_time >= cutoff
                )
                count += sum(
                    1 for r in rep.running
                    if r.spec.arrival_time >= cutoff
                )
            elapsed = min(window, self.sim.now) or 1.0
            return [MetricValue(value=count / elapsed, timestamp=ts)]

        if query in (reg.QUERY_AVG_TTFT, reg.QUERY_AVG_ITL):
            # mean over the last 2m of completed requests (the PromQL
            # rate(sum)/rate(count) histogram-average analog); values in
            # SECONDS like the vllm histograms
            cutoff = self.sim.now - 120.0
            recent = [c for c in model.completed if c.finish_time >= cutoff]
            if not recent:
                return []
            if query == reg.QUERY_AVG_TTFT:
                mean = sum(c.ttft for c in recent) / len(recent)
            else:
                mean = sum(c.itl for c in recent) / len(recent)
            return [MetricValue(value=mean, timestamp=ts)]

        if query == reg.QUERY_MODEL_REQUEST_COUNT:
            retention = parse_go_duration(params.get("retentionPeriod", "10m"))
            cutoff = self.sim.now - retention
            count = sum(1 for c in model.completed if c.finish_time >= cutoff)
            return [MetricValue(value=float(count), timestamp=ts)]

        values: List[MetricValue] = []
        for pod_name, sim in self._pods_of_model(model):
            labels = {"pod": pod_name}
            if query == reg.QUERY_KV_CACHE_USAGE:
                values.append(MetricValue(
                    value=sim.peak_kv_and_reset(), timestamp=ts, labels=labels
                ))
            elif query == reg.QUERY_QUEUE_LENGTH:
                values.append(MetricValue(
                    value=float(sim.peak_queue_and_reset()),
                    timestamp=ts,
                    labels=labels,
                ))
            elif query == reg.QUERY_CACHE_CONFIG_INFO:
                values.append(MetricValue(
                    value=1.0,
                    timestamp=ts,
                    labels={
                        **labels,
                        "num_gpu_blocks": str(sim.profile.num_gpu_blocks),
                        "block_size": str(sim.profile.block_size),
                    },
                ))
            elif query == reg.QUERY_AVG_OUTPUT_TOKENS:
                self._snapshot_counters(pod_name, sim)
                ratio = self._rate_ratio(pod_name, 3, 4)
                if ratio is not None:
                    values.append(MetricValue(
                        value=ratio, timestamp=ts, labels=labels
                    ))
            elif query == reg.QUERY_AVG_INPUT_TOKENS:
                self._snapshot_counters(pod_name, sim)
                ratio = self._rate_ratio(pod_name, 1, 2)
                if ratio is not None:
                    values.append(MetricValue(
                        value=ratio, timestamp=ts, labels=labels
                    ))
            elif query == reg.QUERY_PREFIX_CACHE_HIT_RATE:
                self._snapshot_counters(pod_name, sim)
                hist = self._counter_history.get(pod_name)
                if hist and len(hist) >= 2:
                    newest, oldest = hist[-1], hist[0]
                    dq = newest[6] - oldest[6]
                    dh = newest[5] - oldest[5]
                    if dq > 0:
                        values.append(MetricValue(
                            value=dh / dq, timestamp=ts, labels=labels
                        ))
            else:
                raise KeyError(f"unknown query {query!r}")
        return values
