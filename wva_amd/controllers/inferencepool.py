"""InferencePool reconciler.

Parity: reference internal/controller/inferencepool_reconciler.go:41-118 +
internal/utils/pool/pool.go:34-151 — converts InferencePool (v1
inference.networking.k8s.io or v1alpha2 x-k8s) resources into internal
EndpointPools and stores them in the datastore, which instantiates a
PodScrapingSource per pool. The EPP metrics port is the service port whose
name contains "metric" (pool.go:117), falling back to 9090.
"""
from __future__ import annotations

from typing import Optional

from ..datastore.datastore import Datastore
from ..kube.fake import FakeCluster
from ..kube.objects import EndpointPicker, EndpointPool, InferencePool, Service
from ..utils.logging import get_logger

log = get_logger("controllers.inferencepool")

DEFAULT_EPP_METRICS_PORT = 9090


class InferencePoolReconciler:
    def __init__(self, cluster: FakeCluster, datastore: Datastore):
        self.cluster = cluster
        self.datastore = datastore

    def _metrics_port_for_service(self, namespace: str, service_name: str) -> int:
        svc: Optional[Service] = self.cluster.try_get(
            "Service", namespace, service_name
        )
        if svc is None:
            return DEFAULT_EPP_METRICS_PORT
        for port in svc.ports:
            if "metric" in (port.name or "").lower():
                return port.port
        return DEFAULT_EPP_METRICS_PORT

    def reconcile(self, namespace: str, name: str) -> None:
        pool: Optional[InferencePool] = self.cluster.try_get(
            "InferencePool", namespace, name
        )
        if pool is None:
            self.datastore.pool_delete(namespace, name)
            log.info("removed EndpointPool %s/%s", namespace, name)
            return

        epp_ns = namespace
        epp_service = pool.epp_service_name or f"{name}-epp"
        endpoint_pool = EndpointPool(
            name=name,
            namespace=namespace,
            selector=dict(pool.selector),
            endpoint_picker=EndpointPicker(
                service_name=epp_service,
                namespace=epp_ns,
                metrics_port_number=self._metrics_port_for_service(
                    epp_ns, epp_service
                ),
            ),
        )
        self.datastore.pool_set(endpoint_pool)
        log.info(
            "registered EndpointPool %s/%s (epp service %s)",
            namespace,
            name,
            epp_service,
        )
