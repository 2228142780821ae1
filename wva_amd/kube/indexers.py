"""Field indexers: scaleTargetRef → VariantAutoscaling lookup.

Parity: reference internal/indexers/indexers.go:36-111 — composite key
`ns/apiVersion/Kind/name` gives O(1) VA lookup per Deployment and enforces
at most one VA per scale target.
"""
from __future__ import annotations

from typing import Dict, List, Optional

from ..api.types import VariantAutoscaling
from .fake import FakeCluster


def scale_target_key(
    namespace: str, api_version: str, kind: str, name: str
) -> str:
    return f"{namespace}/{api_version}/{kind}/{name}"


def va_scale_target_key(va: VariantAutoscaling) -> str:
    ref = va.spec.scale_target_ref
    return scale_target_key(va.namespace, ref.api_version, ref.kind, ref.name)


class VAIndex:
    """Live index over the cluster's VariantAutoscalings (recomputed per
    lookup — the FakeCluster is the source of truth and cheap to scan;
    a real informer cache would maintain this incrementally)."""

    def __init__(self, cluster: FakeCluster):
        self.cluster = cluster

    def _index(self) -> Dict[str, List[VariantAutoscaling]]:
        idx: Dict[str, List[VariantAutoscaling]] = {}
        for va in self.cluster.list("VariantAutoscaling"):
            idx.setdefault(va_scale_target_key(va), []).append(va)
        return idx

    def find_va_for_deployment(
        self, namespace: str, name: str, api_version: str = "apps/v1",
        kind: str = "Deployment",
    ) -> Optional[VariantAutoscaling]:
        key = scale_target_key(namespace, api_version, kind, name)
        matches = self._index().get(key, [])
        if not matches:
            return None
        if len(matches) > 1:
            raise ValueError(
                f"multiple VariantAutoscalings reference scale target {key}: "
                f"{[va.name for va in matches]} — at most one is allowed"
            )
        return matches[0]

    def validate_unique_targets(self) -> List[str]:
        """Return error strings for any scale target referenced by >1 VA."""
        errors = []
        for key, matches in self._index().items():
            if len(matches) > 1:
                errors.append(
                    f"scale target {key} referenced by multiple VAs: "
                    f"{sorted(va.name for va in matches)}"
                )
        return errors
