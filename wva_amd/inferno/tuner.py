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
# ... and for 1 DOF (ITL-only updates under saturation)
NIS_THRESHOLD_1DOF = 3.841


@dataclass
class TunerConfig:
    max_batch_size: int = 256
    max_queue_size: int = 2560
    process_noise: float = 1e-2  # Q diagonal
    measurement_noise: float = 4.0  # R diagonal floor (msec² scale)
    # R scales with the observation magnitude: latency measurement error
    # is fractional (scrape-window averaging, mixed batches), not a
    # fixed ms² — an absolute R makes the NIS gate reject everything
    # once TTFT is in the hundreds of ms and the filter can never
    # converge from a badly seeded θ
    relative_noise: float = 0.2
    # initial θ std as a FRACTION of each parameter's seed value —
    # uniform absolute covariance is meaningless across α (tens of ms)
    # and β (hundredths of ms/req); ConfigMap seeds are coarse, so 50%
    initial_rel_std: float = 0.5
    initial_covariance: float = 1.0  # floor for zero-seeded parameters
    nis_threshold: float = NIS_THRESHOLD_2DOF
    max_consecutive_rejections: int = 5
    # covariance inflation per rejected observation (adaptive filtering:
    # persistent innovation ⇒ the state is wrong, widen until the gate
    # re-opens instead of rejecting forever)
    rejection_inflation: float = 1.5
    # trust region: max fractional change of each parameter per accepted
    # observation (EKF linearization validity bound)
    max_step_frac: float = 0.25


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
        rel = self.config.initial_rel_std
        self._P0 = np.diag([
            max((initial.alpha * rel) ** 2, self.config.initial_covariance),
            max((initial.beta * rel) ** 2, 1e-6),
            max((initial.gamma * rel) ** 2, 1e-6),
        ])
        self.P = self._P0.copy()
        # process noise scaled to the per-parameter prior (random-walk
        # drift proportional to each parameter's magnitude)
        self.Q = self._P0 * self.config.process_noise
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

    def update(self, obs: Observation, itl_only: bool = False) -> bool:
        """One EKF step; returns True if the observation was accepted
        (NIS gate), False if rejected as an outlier.

        ``itl_only`` drops the TTFT component of the measurement: when
        the server is saturated (observed rate ≥ the model's max), the
        observed TTFT is dominated by unbounded real backlog that the
        bounded-queue model cannot represent — but the ITL observation
        (α + β·batch at the running batch size) remains valid, so a
        1-D update still corrects the service parameters."""
        try:
            y_pred = self._h(self.theta, obs)
        except (ValueError, ZeroDivisionError):
            return False
        z = np.array([obs.ttft_ms, obs.itl_ms])
        innovation = z - y_pred

        H = self._jacobian(obs)
        P_pred = self.P + self.Q
        # observation-scaled R (see TunerConfig.relative_noise)
        rel = self.config.relative_noise
        R = np.diag([
            max(self.config.measurement_noise, (rel * abs(z[0])) ** 2),
            max(self.config.measurement_noise, (rel * abs(z[1])) ** 2),
        ])
        if itl_only:
            innovation = innovation[1:]
            H = H[1:, :]
            R = R[1:, 1:]
        S = H @ P_pred @ H.T + R
        try:
            S_inv = np.linalg.inv(S)
        except np.linalg.LinAlgError:
            return False

        nis = float(innovation @ S_inv @ innovation)
        self.history.append(nis)

        threshold = NIS_THRESHOLD_1DOF if itl_only else self.config.nis_threshold
        if nis > threshold:
            # outlier: stash current state on first rejection, roll back
            # (unstash) after too many consecutive rejections — the world
            # changed, re-anchor on the pre-outlier estimate and widen P.
            if self._stash is None:
                self._stash = (self.theta.copy(), self.P.copy())
            self._consecutive_rejections += 1
            # widen uncertainty so a persistently surprising world
            # re-opens the gate (otherwise a badly seeded θ is stuck)
            self.P = self.P * self.config.rejection_inflation
            if self._consecutive_rejections >= self.config.max_consecutive_rejections:
                self.theta, _stash_P = self._stash
                self.P = _stash_P * (
                    self.config.rejection_inflation
                    ** self.config.max_consecutive_rejections
                ) + self._P0
                self._stash = None
                self._consecutive_rejections = 0
            return False

        self._stash = None
        self._consecutive_rejections = 0

        K = P_pred @ H.T @ S_inv
        step = K @ innovation
        # trust region: the queueing surface is strongly nonlinear near
        # saturation — an unclamped EKF step overshoots the linearization
        # and oscillates. Bound each parameter's move to a fraction of
        # its magnitude per accepted observation.
        frac = self.config.max_step_frac
        limit = np.maximum(np.abs(self.theta) * frac, [0.5, 1e-3, 1e-3])
        step = np.clip(step, -limit, limit)
        self.theta = self.theta + step
        # parameters are physically non-negative
        self.theta = np.maximum(self.theta, [1e-6, 0.0, 0.0])
        self.P = (np.eye(3) - K @ H) @ P_pred
        return True
