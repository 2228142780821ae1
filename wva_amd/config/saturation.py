"""Saturation-scaling configuration.

Parity: reference internal/interfaces/saturation_scaling.go:8-112 — same
YAML keys (camelCase), same defaults, same validation rules, including the
V2 rules (scaleUpThreshold in (0,1], > scaleDownBoundary) applied only when
analyzerName == "saturation".
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from .. import constants as C

ANALYZER_NAME_V2 = "saturation"
ANALYZER_NAME_INFERNO = "inferno"
# analyzers that run the V2 optimizer flow (token/SLO capacity +
# CostAwareOptimizer) and therefore need the V2 thresholds defaulted —
# the inferno path falls back to the V2 token analyzer when its system
# config is incomplete, so it must carry valid thresholds too
_V2_FLOW_ANALYZERS = (ANALYZER_NAME_V2, ANALYZER_NAME_INFERNO)


class ConfigValidationError(ValueError):
    pass


@dataclass
class SaturationScalingConfig:
    model_id: str = ""  # only used in per-model override entries
    namespace: str = ""  # only used in per-model override entries
    kv_cache_threshold: float = C.DEFAULT_KV_CACHE_THRESHOLD
    queue_length_threshold: float = C.DEFAULT_QUEUE_LENGTH_THRESHOLD
    kv_spare_trigger: float = C.DEFAULT_KV_SPARE_TRIGGER
    queue_spare_trigger: float = C.DEFAULT_QUEUE_SPARE_TRIGGER
    enable_limiter: bool = False
    analyzer_name: str = ""  # "" → V1 percentage, "saturation" → V2 token-based
    scale_up_threshold: float = 0.0
    scale_down_boundary: float = 0.0
    # Fraction of the scheduler (EPP flow-control) queue's token footprint
    # counted as concurrent demand. 1.0 = reference parity (the whole
    # backlog held concurrently, engine_v2.go); < 1.0 models that queued
    # requests drain within the optimization interval — set it to
    # min(1, avg request service time / optimization interval). Improvement
    # over the reference for fast accelerators (MI355X drains hundreds of
    # requests per second per replica).
    scheduler_queue_drain_factor: float = 1.0
    # Predictive scale-up: inflate V2 demand by its observed growth rate
    # times this many seconds (covers pod-ready + engine-tick latency so
    # capacity lands BEFORE the ramp reaches it). 0 disables (reference
    # parity — the reference is purely reactive).
    scale_up_lead_seconds: float = 0.0
    # Per-model override entries (keyed by "modelID|namespace"); populated by
    # the ConfigMap parser from the `overrides` list.
    overrides: Dict[str, "SaturationScalingConfig"] = field(default_factory=dict)

    def get_analyzer_name(self) -> str:
        return self.analyzer_name

    def apply_defaults(self) -> "SaturationScalingConfig":
        if self.analyzer_name in _V2_FLOW_ANALYZERS:
            if self.scale_up_threshold == 0:
                self.scale_up_threshold = C.DEFAULT_SCALE_UP_THRESHOLD
            if self.scale_down_boundary == 0:
                self.scale_down_boundary = C.DEFAULT_SCALE_DOWN_BOUNDARY
        return self

    def validate(self) -> None:
        if not 0 <= self.kv_cache_threshold <= 1:
            raise ConfigValidationError(
                f"kvCacheThreshold must be between 0 and 1, got {self.kv_cache_threshold:.2f}"
            )
        if self.queue_length_threshold < 0:
            raise ConfigValidationError(
                f"queueLengthThreshold must be >= 0, got {self.queue_length_threshold:.1f}"
            )
        if not 0 <= self.kv_spare_trigger <= 1:
            raise ConfigValidationError(
                f"kvSpareTrigger must be between 0 and 1, got {self.kv_spare_trigger:.2f}"
            )
        if self.queue_spare_trigger < 0:
            raise ConfigValidationError(
                f"queueSpareTrigger must be >= 0, got {self.queue_spare_trigger:.1f}"
            )
        if self.kv_cache_threshold < self.kv_spare_trigger:
            raise ConfigValidationError(
                f"kvCacheThreshold ({self.kv_cache_threshold:.2f}) should be >= "
                f"kvSpareTrigger ({self.kv_spare_trigger:.2f})"
            )
        if self.analyzer_name in _V2_FLOW_ANALYZERS:
            if not 0 < self.scale_up_threshold <= 1:
                raise ConfigValidationError(
                    f"scaleUpThreshold must be in (0, 1], got {self.scale_up_threshold:.2f}"
                )
            if not 0 < self.scale_down_boundary <= 1:
                raise ConfigValidationError(
                    f"scaleDownBoundary must be in (0, 1], got {self.scale_down_boundary:.2f}"
                )
            if self.scale_up_threshold <= self.scale_down_boundary:
                raise ConfigValidationError(
                    f"scaleUpThreshold ({self.scale_up_threshold:.2f}) must be > "
                    f"scaleDownBoundary ({self.scale_down_boundary:.2f})"
                )

    def for_model(self, model_id: str, namespace: str) -> "SaturationScalingConfig":
        """Resolve a per-model override if one exists, else this config."""
        return self.overrides.get(f"{model_id}|{namespace}", self)

    # --- YAML wire format (camelCase, ConfigMap `wva-saturation-scaling-config`) ---

    @classmethod
    def from_dict(cls, d: Optional[Dict[str, Any]]) -> "SaturationScalingConfig":
        d = d or {}
        cfg = cls(
            model_id=d.get("model_id", ""),
            namespace=d.get("namespace", ""),
            kv_cache_threshold=float(
                d.get("kvCacheThreshold", C.DEFAULT_KV_CACHE_THRESHOLD)
            ),
            queue_length_threshold=float(
                d.get("queueLengthThreshold", C.DEFAULT_QUEUE_LENGTH_THRESHOLD)
            ),
            kv_spare_trigger=float(d.get("kvSpareTrigger", C.DEFAULT_KV_SPARE_TRIGGER)),
            queue_spare_trigger=float(
                d.get("queueSpareTrigger", C.DEFAULT_QUEUE_SPARE_TRIGGER)
            ),
            enable_limiter=bool(d.get("enableLimiter", False)),
            analyzer_name=d.get("analyzerName", "") or "",
            scale_up_threshold=float(d.get("scaleUpThreshold", 0.0)),
            scale_down_boundary=float(d.get("scaleDownBoundary", 0.0)),
            scheduler_queue_drain_factor=float(
                d.get("schedulerQueueDrainFactor", 1.0)
            ),
            scale_up_lead_seconds=float(d.get("scaleUpLeadSeconds", 0.0)),
        )
        cfg.apply_defaults()
        for entry in d.get("overrides") or []:
            sub = cls.from_dict(entry)
            if sub.model_id:
                cfg.overrides[f"{sub.model_id}|{sub.namespace}"] = sub
        return cfg

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {
            "kvCacheThreshold": self.kv_cache_threshold,
            "queueLengthThreshold": self.queue_length_threshold,
            "kvSpareTrigger": self.kv_spare_trigger,
            "queueSpareTrigger": self.queue_spare_trigger,
        }
        if self.model_id:
            d["model_id"] = self.model_id
        if self.namespace:
            d["namespace"] = self.namespace
        if self.enable_limiter:
            d["enableLimiter"] = True
        if self.analyzer_name:
            d["analyzerName"] = self.analyzer_name
        if self.scale_up_threshold:
            d["scaleUpThreshold"] = self.scale_up_threshold
        if self.scale_down_boundary:
            d["scaleDownBoundary"] = self.scale_down_boundary
        if self.scheduler_queue_drain_factor != 1.0:
            d["schedulerQueueDrainFactor"] = self.scheduler_queue_drain_factor
        if self.scale_up_lead_seconds:
            d["scaleUpLeadSeconds"] = self.scale_up_lead_seconds
        if self.overrides:
            d["overrides"] = [o.to_dict() for o in self.overrides.values()]
        return d
