"""Condition helpers.

Parity: reference api/v1alpha1/conditions.go:9-34 — set_condition updates
ObservedGeneration always and LastTransitionTime only when the status value
changes (meta.SetStatusCondition semantics).
"""
from __future__ import annotations

from typing import Optional

from .types import Condition, VariantAutoscaling, utcnow


def set_condition(
    va: VariantAutoscaling,
    cond_type: str,
    status: str,
    reason: str,
    message: str = "",
) -> None:
    existing = get_condition(va, cond_type)
    if existing is None:
        va.status.conditions.append(
            Condition(
                type=cond_type,
                status=status,
                reason=reason,
                message=message,
                last_transition_time=utcnow(),
                observed_generation=va.metadata.generation,
            )
        )
        return
    if existing.status != status:
        existing.last_transition_time = utcnow()
    existing.status = status
    existing.reason = reason
    existing.message = message
    existing.observed_generation = va.metadata.generation


def get_condition(va: VariantAutoscaling, cond_type: str) -> Optional[Condition]:
    for c in va.status.conditions:
        if c.type == cond_type:
            return c
    return None


def is_condition_true(va: VariantAutoscaling, cond_type: str) -> bool:
    c = get_condition(va, cond_type)
    return c is not None and c.status == "True"


def is_condition_false(va: VariantAutoscaling, cond_type: str) -> bool:
    c = get_condition(va, cond_type)
    return c is not None and c.status == "False"
