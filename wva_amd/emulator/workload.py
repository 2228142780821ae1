"""Workload QPS profiles for the emulator and bench.

BASELINE configs use a constant-QPS plumbing test and a synthetic QPS ramp
(the north-star scenario). Profiles are functions t_seconds → requests/s.
"""
from __future__ import annotations

from typing import Callable, List, Tuple

QPSProfile = Callable[[float], float]


def constant_qps(qps: float) -> QPSProfile:
    return lambda t: qps


def ramp_qps(stages: List[Tuple[float, float]]) -> QPSProfile:
    """Piecewise-constant ramp: stages = [(duration_s, qps), ...]."""

    def profile(t: float) -> float:
        acc = 0.0
        for duration, qps in stages:
            acc += duration
            if t < acc:
                return qps
        return stages[-1][1] if stages else 0.0

    return profile


def burst_qps(base: float, burst: float, period: float, duty: float = 0.2) -> QPSProfile:
    """Square-wave burst load (burst_load_generator.sh analog)."""

    def profile(t: float) -> float:
        phase = (t % period) / period
        return burst if phase < duty else base

    return profile
