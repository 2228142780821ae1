"""Online (α, β, γ) estimation with an Extended Kalman Filter.

Parity: reference internal/engines/analyzers/queueingmodel/tuner (which
uses the external llm-inferno/kalman-filter + gonum): the state is the
ServiceParms vector θ = (α, β, γ); the observation function h(θ) solves
the queueing model at the observed request rate and returns (TTFT, ITL);
the EKF updates θ from measured (TTFT, ITL) with NIS (normalized
innovation squared) outlier rejection and stash/unstash rollback.

numpy implementation; the Jacobian is computed by central finite
differences of h(θ) (the reference's kalman-filter library does the same
numerically).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import numpy as np

from .queue_analyzer import Configuration, QueueAnalyzer, RequestSize, ServiceParms

# Chi-squared 95th percentile for 2 degrees of freedom (TTFT, ITL)
NIS_THRESHOLD_2DOF = 5.991


@dataclass
class TunerConfig:
    max_batch_size: int = 256
    max_queue_size: int = 2560
    process_noise: float = 1e-4  # Q diagonal
    measurement_noise: float = 4.0  # R diagonal (msec² scale)
    initial_covariance: float = 1.0
    nis_threshold: float = NIS_THRESHOLD_2DOF
    max_consecutive_rejections: int = 5


@dataclass
class Observation:
    request_rate: float  # requests/sec
    avg_input_tokens: float
    avg_output_tokens: float
    ttft_ms: float
    itl_ms: float


class ServiceParmsTuner:
    """EKF over θ = (α, β, γ)."""

    def __init__(self, initial: ServiceParms, config: Optional[TunerConfig] = None):
        self.config = config or TunerConfig()
        self.theta = np.array(
            [initial.alpha, initial.beta, initial.gamma], dtype=np.float64
        )
        self.P = np.eye(3) * self.config.initial_covariance
        self.Q = np.eye(3) * self.config.process_noise
        self.R = np.eye(2) * self.config.measurement_noise
        self._stash: Optional[Tuple[np.ndarray, np.ndarray]] = None
        self._consecutive_rejections = 0
        self.history: List[float] = []  # NIS values

    def parms(self) -> ServiceParms:
        return ServiceParms(
            alpha=float(self.theta[0]),
            beta=float(self.theta[1]),
            gamma=float(self.theta[2]),
        )

    # --- observation model ---

    def _h(self, theta: np.ndarray, obs: Observation) -> np.ndarray:
        """Predicted (TTFT, ITL) by solving the queueing model at the
        observed rate with candidate parameters."""
        parms = ServiceParms(
            alpha=max(float(theta[0]), 1e-6),
            beta=max(float(theta[1]), 0.0),
            gamma=max(float(theta[2]), 0.0),
        )
        analyzer = QueueAnalyzer(
            Configuration(
                max_batch_size=self.config.max_batch_size,
                max_queue_size=self.config.max_queue_size,
                service_parms=parms,
            ),
            RequestSize(
                avg_input_tokens=obs.avg_input_tokens,
                avg_output_tokens=obs.avg_output_tokens,
            ),
        )
        rate = min(obs.request_rate, analyzer.rate_max * 0.999)
        rate = max(rate, analyzer.rate_min)
        m = analyzer.analyze(rate)
        return np.array([m.avg_ttft, m.avg_token_time])

    def _jacobian(self, obs: Observation) -> np.ndarray:
        H = np.zeros((2, 3))
        eps_scale = np.maximum(np.abs(self.theta) * 1e-3, 1e-6)
        for j in range(3):
            dt = np.zeros(3)
            dt[j] = eps_scale[j]
            try:
                y_plus = self._h(self.theta + dt, obs)
                y_minus = self._h(self.theta - dt, obs)
            except (ValueError, ZeroDivisionError):
                continue
            H[:, j] = (y_plus - y_minus) / (2 * eps_scale[j])
        return H

    # --- update ---

    def update(self, obs: Observation) -> bool:
        """One EKF step; returns True if the observation was accepted
        (NIS gate), False if rejected as an outlier."""
        try:
            y_pred = self._h(self.theta, obs)
        except (ValueError, ZeroDivisionError):
            return False
        z = np.array([obs.ttft_ms, obs.itl_ms])
        innovation = z - y_pred

        H = self._jacobian(obs)
        P_pred = self.P + self.Q
        S = H @ P_pred @ H.T + self.R
        try:
            S_inv = np.linalg.inv(S)
        except np.linalg.LinAlgError:
            return False

        nis = float(innovation @ S_inv @ innovation)
        self.history.append(nis)

        if nis > self.config.nis_threshold:
            # outlier: stash current state on first rejection, roll back
            # (unstash) after too many consecutive rejections — the world
            # changed, re-anchor on the pre-outlier estimate and widen P.
            if self._stash is None:
                self._stash = (self.theta.copy(), self.P.copy())
            self._consecutive_rejections += 1
            if self._consecutive_rejections >= self.config.max_consecutive_rejections:
                self.theta, self.P = self._stash
                self.P = self.P + np.eye(3) * self.config.initial_covariance
                self._stash = None
                self._consecutive_rejections = 0
            return False

        self._stash = None
        self._consecutive_rejections = 0

        K = P_pred @ H.T @ S_inv
        self.theta = self.theta + K @ innovation
        # parameters are physically non-negative
        self.theta = np.maximum(self.theta, [1e-6, 0.0, 0.0])
        self.P = (np.eye(3) - K @ H) @ P_pred
        return True
