"""Queue analyzer: service-rate construction + SLO sizing.

Parity: reference pkg/analyzer/queueanalyzer.go —
  iterationTime(n) = α + n·(β·tokensCompute + γ·tokensMemory)
    tokensCompute = (I + O) / (O + 1);  tokensMemory = I + O/2
  prefillTime(n)  = iterationTime(n) + (β+γ)·I
  decodeTime(n)   = iterationTime(n) + β + γ·(I + O/2)
  servRate[n] = n / (prefillTime(n) + O·decodeTime(n))   (requests/msec)
  Size(targets) → max rates via binary search on monotone TTFT/ITL curves,
  TPS capped at (1 − StabilitySafetyFraction)·λmax.

On MI355X the (α, β, γ) come from wva_amd.calibration measurements — β
(compute slope) maps to MFMA-bound decode FLOPs, γ (memory slope) to the
HBM3E-bound KV/weight streaming; per-TP-degree fits capture the RCCL/xGMI
all-reduce term in α.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Tuple

from .queueing import MM1StateDependentModel, binary_search

EPSILON = 0.001
STABILITY_SAFETY_FRACTION = 0.1
DEFAULT_MAX_NUM_TOKENS = 8192


@dataclass
class ServiceParms:
    alpha: float = 0.0  # base (msec)
    beta: float = 0.0  # slope for compute time
    gamma: float = 0.0  # slope for memory access time

    def iteration_time(self, r: "RequestSize", batch_size: float) -> float:
        tokens_compute = (r.avg_input_tokens + r.avg_output_tokens) / (
            r.avg_output_tokens + 1
        )
        tokens_memory = r.avg_input_tokens + r.avg_output_tokens / 2
        return self.alpha + batch_size * (
            self.beta * tokens_compute + self.gamma * tokens_memory
        )

    def prefill_time(self, r: "RequestSize", batch_size: float) -> float:
        if r.avg_input_tokens == 0:
            return 0.0
        return self.iteration_time(r, batch_size) + (
            (self.beta + self.gamma) * r.avg_input_tokens
        )

    def decode_time(self, r: "RequestSize", batch_size: float) -> float:
        return (
            self.iteration_time(r, batch_size)
            + self.beta
            + self.gamma * (r.avg_input_tokens + r.avg_output_tokens / 2)
        )


@dataclass
class RequestSize:
    avg_input_tokens: float = 0.0
    avg_output_tokens: float = 0.0

    def check(self) -> None:
        if self.avg_input_tokens < 0 or self.avg_output_tokens < 0:
            raise ValueError("token counts must be non-negative")


@dataclass
class Configuration:
    max_batch_size: int = 0
    max_num_tokens: int = DEFAULT_MAX_NUM_TOKENS
    max_queue_size: int = 0
    service_parms: Optional[ServiceParms] = None

    def check(self) -> None:
        if self.max_batch_size <= 0:
            raise ValueError("max_batch_size must be > 0")
        if self.max_queue_size < 0:
            raise ValueError("max_queue_size must be >= 0")
        if self.service_parms is None:
            raise ValueError("service_parms required")


@dataclass
class AnalysisMetrics:
    throughput: float = 0.0  # requests/sec
    avg_resp_time: float = 0.0  # msec
    avg_wait_time: float = 0.0  # msec
    avg_num_in_serv: float = 0.0
    avg_prefill_time: float = 0.0  # msec
    avg_token_time: float = 0.0  # ITL, msec
    avg_ttft: float = 0.0  # msec
    max_rate: float = 0.0  # requests/sec
    rho: float = 0.0


@dataclass
class TargetPerf:
    target_ttft: float = 0.0  # msec (queueing + prefill)
    target_itl: float = 0.0  # msec
    target_tps: float = 0.0  # tokens/sec

    def check(self) -> None:
        if self.target_ttft < 0 or self.target_itl < 0 or self.target_tps < 0:
            raise ValueError("targets must be non-negative")
        if self.target_ttft == 0 and self.target_itl == 0 and self.target_tps == 0:
            raise ValueError("at least one target must be set")


@dataclass
class TargetRate:
    rate_target_ttft: float = 0.0  # requests/sec
    rate_target_itl: float = 0.0
    rate_target_tps: float = 0.0


class QueueAnalyzer:
    def __init__(self, config: Configuration, request_size: RequestSize):
        config.check()
        request_size.check()
        self.max_batch_size = config.max_batch_size
        self.max_num_tokens = config.max_num_tokens
        self.max_queue_size = config.max_queue_size
        self.service_parms = config.service_parms
        self.request_size = request_size

        parms, r = self.service_parms, self.request_size
        serv_rate = []
        for n in range(1, config.max_batch_size + 1):
            prefill = parms.prefill_time(r, n)
            decode = r.avg_output_tokens * parms.decode_time(r, n)
            serv_rate.append(n / (prefill + decode))
        self.serv_rate = serv_rate
        lam_min = serv_rate[0] * EPSILON
        lam_max = serv_rate[-1] * (1 - EPSILON)
        self.rate_min = lam_min * 1000.0  # per-second
        self.rate_max = lam_max * 1000.0
        occupancy = config.max_queue_size + config.max_batch_size
        self.model = MM1StateDependentModel(occupancy, serv_rate)

    # --- evaluation ---

    def _solve(self, lam_per_ms: float) -> None:
        self.model.solve(lam_per_ms, 1.0)
        if not self.model.is_valid:
            raise ValueError("invalid queueing model state")

    def _ttft_itl_at(self, lam_per_ms: float) -> Tuple[float, float]:
        self._solve(lam_per_ms)
        avg_in_serv = self.model.avg_num_in_servers
        prefill = self.service_parms.prefill_time(self.request_size, avg_in_serv)
        decode = (self.model.avg_serv_time - prefill) / (
            self.request_size.avg_output_tokens
        )
        ttft = self.model.avg_wait_time + prefill + decode
        return ttft, decode

    def analyze(self, request_rate: float) -> AnalysisMetrics:
        """Performance at request_rate (requests/sec)."""
        if request_rate <= 0:
            raise ValueError(f"invalid request rate {request_rate}")
        if request_rate > self.rate_max:
            raise ValueError(
                f"rate={request_rate}, max allowed rate={self.rate_max}"
            )
        self._solve(request_rate / 1000.0)
        m = self.model
        avg_in_serv = m.avg_num_in_servers
        prefill = self.service_parms.prefill_time(self.request_size, avg_in_serv)
        decode = (m.avg_serv_time - prefill) / self.request_size.avg_output_tokens
        ttft = m.avg_wait_time + prefill + decode
        rho = min(max(avg_in_serv / self.max_batch_size, 0.0), 1.0)
        return AnalysisMetrics(
            throughput=m.throughput * 1000.0,
            avg_resp_time=m.avg_resp_time,
            avg_wait_time=m.avg_wait_time,
            avg_num_in_serv=avg_in_serv,
            avg_prefill_time=prefill,
            avg_token_time=decode,
            avg_ttft=ttft,
            max_rate=self.rate_max,
            rho=rho,
        )

    def size(
        self, target: TargetPerf
    ) -> Tuple[TargetRate, AnalysisMetrics, TargetPerf]:
        """Max request rates meeting each target; returns (rates, metrics at
        the min rate, achieved targets). Raises when a target is below the
        feasible region."""
        target.check()
        lam_min = self.rate_min / 1000.0
        lam_max = self.rate_max / 1000.0

        lam_ttft = lam_max
        if target.target_ttft > 0:
            lam_ttft, ind = binary_search(
                lam_min, lam_max, target.target_ttft,
                lambda x: self._ttft_itl_at(x)[0],
            )
            if ind < 0:
                raise ValueError("TTFT target below the bounded region")

        lam_itl = lam_max
        if target.target_itl > 0:
            lam_itl, ind = binary_search(
                lam_min, lam_max, target.target_itl,
                lambda x: self._ttft_itl_at(x)[1],
            )
            if ind < 0:
                raise ValueError("ITL target below the bounded region")

        lam_tps = lam_max
        if target.target_tps > 0:
            lam_tps = lam_max * (1 - STABILITY_SAFETY_FRACTION)

        lam = min(lam_ttft, lam_itl, lam_tps)
        metrics = self.analyze(lam * 1000.0)
        rates = TargetRate(
            rate_target_ttft=lam_ttft * 1000.0,
            rate_target_itl=lam_itl * 1000.0,
            rate_target_tps=lam_tps * 1000.0,
        )
        achieved = TargetPerf(
            target_ttft=metrics.avg_ttft,
            target_itl=metrics.avg_token_time,
            target_tps=metrics.throughput * self.request_size.avg_output_tokens,
        )
        return rates, metrics, achieved
