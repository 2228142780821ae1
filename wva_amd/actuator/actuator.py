"""Actuator — metric emission toward HPA/KEDA.

Parity: reference internal/actuator/actuator.go:21-104 — reads the live
Deployment for the real current replica count (status → spec → 1 fallback),
then emits wva_current/desired/ratio gauges; emission failures never fail
the tick.
"""
from __future__ import annotations


from ..api.types import VariantAutoscaling
from ..kube.fake import FakeCluster, NotFoundError
from ..metrics.metrics import MetricsEmitter
from ..utils.backoff import retry_with_backoff
from ..utils.logging import get_logger

log = get_logger("actuator")


class Actuator:
    def __init__(self, cluster: FakeCluster, emitter: MetricsEmitter):
        self.cluster = cluster
        self.emitter = emitter

    def get_current_deployment_replicas(self, va: VariantAutoscaling) -> int:
        def fetch():
            return self.cluster.get(
                "Deployment", va.namespace, va.get_scale_target_name()
            )

        deploy = retry_with_backoff(fetch, retry_on=NotFoundError, max_attempts=2)
        # status first (actual state); a zero status with a non-zero spec
        # means the controller hasn't caught up — use spec, matching the
        # engine's BuildVariantStates fallback (engine.go:525-528)
        if deploy.status.replicas > 0:
            return deploy.status.replicas
        if deploy.replicas:
            return deploy.replicas
        if deploy.status.replicas == 0 and deploy.replicas == 0:
            return 0
        return 1

    def emit_metrics(self, va: VariantAutoscaling) -> None:
        desired = va.status.desired_optimized_alloc.num_replicas
        if desired < 0:
            return
        try:
            current = self.get_current_deployment_replicas(va)
        except Exception as e:  # noqa: BLE001
            log.error(
                "could not get current deployment replicas for %s: %s — using 0",
                va.full_name(),
                e,
            )
            current = 0
        try:
            self.emitter.emit_replica_metrics(
                va.name,
                va.namespace,
                current,
                desired,
                va.status.desired_optimized_alloc.accelerator,
            )
        except Exception as e:  # noqa: BLE001 — metric failures never break the loop
            log.error("failed to emit replica metrics for %s: %s", va.full_name(), e)
