"""Scale-to-zero + minimum-replica enforcer.

Parity: reference internal/engines/pipeline/enforcer.go:18-183 —
  * scale-to-zero enabled: query request count over the retention window;
    0 requests → all targets 0; query error → keep targets (fail safe)
  * disabled: ensure >= 1 total replica, preserved on the cheapest variant
    (alphabetical tie-break, default cost 10.0 when unknown)
"""
from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple

from ..analyzers.interfaces import VariantSaturationAnalysis
from ..config.scale_to_zero import (
    ScaleToZeroConfigData,
    is_scale_to_zero_enabled,
    scale_to_zero_retention_seconds,
)
from ..constants import DEFAULT_VARIANT_COST
from ..utils.logging import get_logger

log = get_logger("pipeline.enforcer")

# (model_id, namespace, retention_seconds) -> request count; raises on error
RequestCountFunc = Callable[[str, str, float], float]


class Enforcer:
    def __init__(self, request_count_func: RequestCountFunc):
        self._request_count_func = request_count_func

    def enforce_policy(
        self,
        model_id: str,
        namespace: str,
        saturation_targets: Dict[str, int],
        variant_analyses: List[VariantSaturationAnalysis],
        scale_to_zero_config: ScaleToZeroConfigData,
    ) -> Tuple[Dict[str, int], bool]:
        """Returns (targets, applied) — applied True when scale-to-zero
        zeroed the targets or min-replica preservation kicked in."""
        if is_scale_to_zero_enabled(scale_to_zero_config, model_id):
            return self._apply_scale_to_zero(
                model_id, namespace, saturation_targets, scale_to_zero_config
            )
        return self._ensure_minimum_replicas(
            model_id, saturation_targets, variant_analyses
        )

    def _apply_scale_to_zero(
        self,
        model_id: str,
        namespace: str,
        targets: Dict[str, int],
        scale_to_zero_config: ScaleToZeroConfigData,
    ) -> Tuple[Dict[str, int], bool]:
        retention = scale_to_zero_retention_seconds(scale_to_zero_config, model_id)
        try:
            request_count = self._request_count_func(model_id, namespace, retention)
        except Exception as e:  # noqa: BLE001 — fail safe: never scale to zero on error
            log.error(
                "failed to get request count, keeping current targets: "
                "model=%s ns=%s err=%s",
                model_id,
                namespace,
                e,
            )
            return targets, False
        if request_count > 0:
            return targets, False
        log.info(
            "no requests in retention period (%.0fs), scaling %s/%s to zero",
            retention,
            namespace,
            model_id,
        )
        for variant in targets:
            targets[variant] = 0
        return targets, True

    def _ensure_minimum_replicas(
        self,
        model_id: str,
        targets: Dict[str, int],
        variant_analyses: List[VariantSaturationAnalysis],
    ) -> Tuple[Dict[str, int], bool]:
        if sum(targets.values()) > 0:
            return targets, False
        costs = {va.variant_name: va.cost for va in variant_analyses}
        cheapest: Optional[str] = None
        cheapest_cost = -1.0
        for variant in targets:
            cost = costs.get(variant, DEFAULT_VARIANT_COST)
            if (
                cheapest_cost < 0
                or cost < cheapest_cost
                or (cost == cheapest_cost and variant < cheapest)
            ):
                cheapest = variant
                cheapest_cost = cost
        if cheapest is not None:
            targets[cheapest] = 1
            log.info(
                "preserving minimum replica on cheapest variant %s (model=%s)",
                cheapest,
                model_id,
            )
            return targets, True
        return targets, False
