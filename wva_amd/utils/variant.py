"""VariantAutoscaling helpers: active/inactive listing, model grouping.

Parity: reference internal/utils/variant.go:38-226 — a VA is "active" when
its target deployment has replicas > 0 and "inactive" at 0 (scale-from-zero
candidates); grouping key is `modelID|namespace`; multi-controller
filtering honors the `wva.llmd.ai/controller-instance` label against the
CONTROLLER_INSTANCE env and the `wva.llmd.ai/exclude` namespace annotation.
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional

from ..api.types import VariantAutoscaling
from ..constants import (
    CONTROLLER_INSTANCE_LABEL_KEY,
    NAMESPACE_EXCLUDE_ANNOTATION_KEY,
)
from ..kube.fake import FakeCluster
from ..kube.objects import Deployment


def controller_instance() -> str:
    return os.environ.get("CONTROLLER_INSTANCE", "")


def matches_controller_instance(va: VariantAutoscaling) -> bool:
    """Multi-controller isolation (predicates.go:184-243): with
    CONTROLLER_INSTANCE set, only VAs labeled for this instance match;
    without it, only unlabeled VAs match."""
    instance = controller_instance()
    label = va.metadata.labels.get(CONTROLLER_INSTANCE_LABEL_KEY, "")
    if instance:
        return label == instance
    return label == ""


def namespace_excluded(cluster: FakeCluster, namespace: str) -> bool:
    ns_obj = cluster.try_get("Namespace", "", namespace)
    if ns_obj is None:
        return False
    return (
        ns_obj.metadata.annotations.get(NAMESPACE_EXCLUDE_ANNOTATION_KEY, "")
        == "true"
    )


def _deployment_for(
    cluster: FakeCluster, va: VariantAutoscaling
) -> Optional[Deployment]:
    return cluster.try_get("Deployment", va.namespace, va.get_scale_target_name())


def list_variant_autoscalings(
    cluster: FakeCluster, namespace: Optional[str] = None
) -> List[VariantAutoscaling]:
    vas = cluster.list("VariantAutoscaling", namespace=namespace)
    return [
        va
        for va in vas
        if matches_controller_instance(va)
        and not namespace_excluded(cluster, va.namespace)
        and va.metadata.deletion_timestamp is None
    ]


def active_variant_autoscalings(
    cluster: FakeCluster, namespace: Optional[str] = None
) -> List[VariantAutoscaling]:
    """VAs whose deployment currently has replicas > 0."""
    out = []
    for va in list_variant_autoscalings(cluster, namespace):
        deploy = _deployment_for(cluster, va)
        if deploy is not None and deploy.replicas > 0:
            out.append(va)
    return out


def inactive_variant_autoscalings(
    cluster: FakeCluster, namespace: Optional[str] = None
) -> List[VariantAutoscaling]:
    """VAs whose deployment is scaled to 0 (scale-from-zero candidates)."""
    out = []
    for va in list_variant_autoscalings(cluster, namespace):
        deploy = _deployment_for(cluster, va)
        if deploy is not None and deploy.replicas == 0:
            out.append(va)
    return out


def group_variant_autoscaling_by_model(
    vas: List[VariantAutoscaling],
) -> Dict[str, List[VariantAutoscaling]]:
    """Group by `modelID|namespace` (variant.go:64-75)."""
    groups: Dict[str, List[VariantAutoscaling]] = {}
    for va in vas:
        groups.setdefault(f"{va.spec.model_id}|{va.namespace}", []).append(va)
    return groups
