"""DirectActuator — scale-subresource writer.

Parity: reference internal/actuator/direct_actuator.go:37-121 — the only
component that writes spec.replicas directly; used exclusively by the
scale-from-zero engine for the 0→1 transition.
"""
from __future__ import annotations

from ..kube.fake import FakeCluster
from ..utils.logging import get_logger

log = get_logger("actuator.direct")


class DirectActuator:
    def __init__(self, cluster: FakeCluster):
        self.cluster = cluster

    def scale_target_object(
        self, kind: str, namespace: str, name: str, replicas: int
    ) -> None:
        self.cluster.scale(kind, namespace, name, replicas)
        log.info("scaled %s %s/%s to %d replicas", kind, namespace, name, replicas)
