"""Watch-event predicates.

Parity: reference internal/controller/predicates.go:31-243 —
  * VariantAutoscalingPredicate: namespace exclusion annotation + multi-
    controller isolation by instance label
  * EventFilter: VA create-only (periodic engine loop handles drift; update
    and delete events are dropped), ConfigMap updates allowed, Deployment
    create+delete only
  * ConfigMapPredicate: well-known names in the controller namespace
    (global) or tracked/opted-in namespaces
"""
from __future__ import annotations

from typing import Callable

from ..api.types import VariantAutoscaling
from ..constants import (
    ACCELERATOR_CONFIG_MAP_NAME,
    MODEL_PERF_CONFIG_MAP_NAME,
    SATURATION_CONFIG_MAP_NAME,
    SCALE_TO_ZERO_CONFIG_MAP_NAME,
    SERVICE_CLASS_CONFIG_MAP_NAME,
    WVA_CONFIG_MAP_NAME,
)
from ..kube.fake import ADDED, DELETED, MODIFIED, FakeCluster, WatchEvent
from ..utils.variant import matches_controller_instance, namespace_excluded

Predicate = Callable[[WatchEvent], bool]

WELL_KNOWN_CONFIGMAPS = {
    WVA_CONFIG_MAP_NAME,
    SATURATION_CONFIG_MAP_NAME,
    SCALE_TO_ZERO_CONFIG_MAP_NAME,
    SERVICE_CLASS_CONFIG_MAP_NAME,
    ACCELERATOR_CONFIG_MAP_NAME,
    MODEL_PERF_CONFIG_MAP_NAME,
}


def variant_autoscaling_predicate(cluster: FakeCluster) -> Predicate:
    def pred(event: WatchEvent) -> bool:
        if event.kind != "VariantAutoscaling":
            return False
        va = event.obj
        assert isinstance(va, VariantAutoscaling)
        if namespace_excluded(cluster, va.namespace):
            return False
        if not matches_controller_instance(va):
            return False
        # VA create (and delete, for namespace untracking) — updates are
        # handled by the periodic engine loop (EventFilter:99-143)
        return event.type in (ADDED, DELETED)

    return pred


def deployment_predicate() -> Predicate:
    def pred(event: WatchEvent) -> bool:
        return event.kind == "Deployment" and event.type in (ADDED, DELETED)

    return pred


def configmap_predicate() -> Predicate:
    def pred(event: WatchEvent) -> bool:
        if event.kind != "ConfigMap":
            return False
        return event.obj.metadata.name in WELL_KNOWN_CONFIGMAPS

    return pred


def inferencepool_predicate() -> Predicate:
    def pred(event: WatchEvent) -> bool:
        return event.kind == "InferencePool" and event.type in (
            ADDED,
            MODIFIED,
            DELETED,
        )

    return pred
