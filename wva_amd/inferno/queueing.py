"""Queueing models: M/M/1/K and state-dependent M/M/1/K + binary search.

Parity: reference pkg/analyzer/{queuemodel,mm1kmodel,
mm1modelstatedependent,utils}.go — same state-probability recursions
(including the overflow-rescaled recursion for the state-dependent model),
Little's-law statistics, and the monotone binary search with boundary
indicator.
"""
from __future__ import annotations

import math
from typing import Callable, List, Sequence, Tuple

EPSILON = 1e-6
MAX_ITERATIONS = 100


def within_tolerance(x: float, value: float, tolerance: float) -> bool:
    if x == value:
        return True
    if value == 0 or tolerance < 0:
        return False
    return abs((x - value) / value) <= tolerance


def binary_search(
    x_min: float,
    x_max: float,
    y_target: float,
    eval_fn: Callable[[float], float],
) -> Tuple[float, int]:
    """Find x* in [x_min, x_max] with eval_fn(x*) = y_target for a
    monotone eval_fn. Returns (x*, indicator) with indicator -1/0/+1 for
    target below/within/above the bounded region. Raises on evaluation
    failure or non-convergence."""
    if x_min > x_max:
        raise ValueError(f"invalid range [{x_min}, {x_max}]")
    y_lo = eval_fn(x_min)
    if within_tolerance(y_lo, y_target, EPSILON):
        return x_min, 0
    y_hi = eval_fn(x_max)
    if within_tolerance(y_hi, y_target, EPSILON):
        return x_max, 0

    increasing = y_lo < y_hi
    if (increasing and y_target < y_lo) or (not increasing and y_target > y_lo):
        return x_min, -1
    if (increasing and y_target > y_hi) or (not increasing and y_target < y_hi):
        return x_max, +1

    lo, hi = x_min, x_max
    x_star = lo
    for _ in range(MAX_ITERATIONS):
        x_star = 0.5 * (lo + hi)
        y_star = eval_fn(x_star)
        if within_tolerance(y_star, y_target, EPSILON):
            return x_star, 0
        if (y_star < y_target) == increasing:
            lo = x_star
        else:
            hi = x_star
        if within_tolerance(lo, hi, EPSILON):
            return x_star, 0
    return x_star, 0


class MM1KModel:
    """M/M/1/K finite-capacity queue."""

    def __init__(self, K: int):
        if K <= 0:
            raise ValueError("K must be positive")
        self.K = K
        self.p: List[float] = [0.0] * (K + 1)
        self.lam = 0.0
        self.mu = 0.0
        self.rho = 0.0
        self.is_valid = False
        self.throughput = 0.0
        self.avg_num_in_system = 0.0
        self.avg_resp_time = 0.0
        self.avg_serv_time = 0.0
        self.avg_wait_time = 0.0
        self.avg_queue_length = 0.0

    # overridable pieces (state-dependent subclass replaces these)
    def _compute_rho(self) -> float:
        if self.lam == self.mu:
            return 1.0
        return self.lam / self.mu

    def _rho_max(self) -> float:
        return float(self.K)

    def solve(self, lam: float, mu: float) -> None:
        self.lam = lam
        self.mu = mu
        if lam < 0 or mu <= 0:
            self.is_valid = False
            return
        self.rho = self._compute_rho()
        if self.rho < 0 or self.rho >= self._rho_max():
            self.is_valid = False
            return
        self.is_valid = True
        self._compute_statistics()

    def _compute_probabilities(self) -> None:
        """Geometric p_i ∝ rho^i, anchored at the distribution's dominant
        end so no intermediate overflows: for rho ≤ 1 work up from p_0,
        for rho > 1 down from p_K with (1/rho)^(K-i) (rho^(K+1) overflows
        float64 once rho^(K+1) > 1e308, e.g. rho≈40 at K≈190 — found by
        the hypothesis suite). Explicit normalization also removes the
        cancellation of (1-rho)/(1-rho^(K+1)) near rho = 1."""
        K, rho = self.K, self.rho
        if rho <= 1:
            q = [rho**i for i in range(K + 1)]
        else:
            r_inv = 1.0 / rho
            q = [r_inv ** (K - i) for i in range(K + 1)]
        total = sum(q)
        for i in range(K + 1):
            self.p[i] = q[i] / total

    def _compute_statistics(self) -> None:
        if not self.is_valid:
            return
        self._compute_probabilities()
        self.avg_num_in_system = sum(i * pi for i, pi in enumerate(self.p))
        self.throughput = self.lam * (1 - self.p[self.K])
        self.avg_resp_time = (
            self.avg_num_in_system / self.throughput if self.throughput else 0.0
        )
        self.avg_serv_time = 1.0 / self.mu
        self.avg_wait_time = max(self.avg_resp_time - self.avg_serv_time, 0.0)
        self.avg_queue_length = self.throughput * self.avg_wait_time


class MM1StateDependentModel(MM1KModel):
    """M/M/1/K with state-dependent service rates serv_rate[n]
    (batch-size-dependent decode rates)."""

    def __init__(self, K: int, serv_rate: Sequence[float]):
        super().__init__(K)
        if not serv_rate:
            raise ValueError("serv_rate must be non-empty")
        self.serv_rate = list(serv_rate)
        self.avg_num_in_servers = 0.0

    def _compute_rho(self) -> float:
        return 1.0 - self.p[0] if self.p else 0.0

    def _rho_max(self) -> float:
        # state-dependent validity: any lam >= 0 with finite K is solvable;
        # rho computed post-hoc from p[0]. Mirror the reference: rho < rhoMax
        # check uses the base class's K bound via probabilities below.
        return float("inf")

    def solve(self, lam: float, mu: float = 1.0) -> None:
        self.lam = lam
        self.mu = mu
        if lam < 0 or mu <= 0:
            self.is_valid = False
            return
        self.is_valid = True
        self._compute_statistics()
        self.rho = self._compute_rho()

    def _compute_probabilities(self) -> None:
        """Birth-death recursion p[n+1] = p[n]·λ/servRate[n] with overflow
        rescaling (mm1modelstatedependent.go:70-113)."""
        K = self.K
        num = len(self.serv_rate)
        scale = 1.7976931348623157e308 / K  # MaxFloat64 / K
        self.p = [0.0] * (K + 1)
        self.p[0] = 1.0
        for n in range(K):
            s_rate = self.serv_rate[n] if n < num else self.serv_rate[-1]
            nxt = self.p[n] * self.lam / s_rate
            while nxt < 0 or math.isinf(nxt) or math.isnan(nxt):
                for i in range(n + 1):
                    self.p[i] /= scale
                nxt = self.p[n] * self.lam / s_rate
            self.p[n + 1] = nxt
        total = 0.0
        for n in range(K + 1):
            total += self.p[n]
            if total < 0 or math.isinf(total):
                total = 0.0
                for i in range(K + 1):
                    self.p[i] /= scale
                    if i <= n:
                        total += self.p[i]
        for n in range(K + 1):
            self.p[n] /= total

    def _compute_statistics(self) -> None:
        if not self.is_valid:
            return
        self._compute_probabilities()
        num = len(self.serv_rate)
        avg_in_servers = 0.0
        avg_in_system = 0.0
        sum_p = self.p[0]
        for i in range(1, self.K + 1):
            avg_in_system += i * self.p[i]
            sum_p += self.p[i]
            if i == num:
                avg_in_servers = avg_in_system + (1 - sum_p) * num
        if self.K <= num:
            avg_in_servers = avg_in_system
        self.avg_num_in_servers = avg_in_servers
        self.avg_num_in_system = avg_in_system
        self.throughput = self.lam * (1 - self.p[self.K])
        if self.throughput > 0:
            self.avg_resp_time = self.avg_num_in_system / self.throughput
            self.avg_serv_time = self.avg_num_in_servers / self.throughput
        else:
            self.avg_resp_time = 0.0
            self.avg_serv_time = 0.0
        self.avg_wait_time = max(self.avg_resp_time - self.avg_serv_time, 0.0)
        self.avg_queue_length = self.throughput * self.avg_wait_time
