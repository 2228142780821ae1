"""WVA output metrics — the HPA/KEDA contract.

Parity: reference internal/metrics/metrics.go:37-165 — byte-identical metric
names and label sets:
  wva_replica_scaling_total{variant_name,namespace,direction,reason[,controller_instance]}
  wva_desired_replicas / wva_current_replicas / wva_desired_ratio
      {variant_name,namespace,accelerator_type[,controller_instance]}
Ratio semantics: current == 0 → ratio emitted as desired (0→N handled as N).

Uses prometheus_client; metrics register into a caller-provided registry so
tests can isolate, and the default REGISTRY in production.
"""
from __future__ import annotations

import os
import threading
from typing import Optional

from prometheus_client import Counter, Gauge, CollectorRegistry, REGISTRY

from .. import constants as C

_lock = threading.Lock()
_emitter: Optional["MetricsEmitter"] = None


class MetricsEmitter:
    def __init__(
        self,
        registry: Optional[CollectorRegistry] = None,
        controller_instance: Optional[str] = None,
    ):
        self.registry = registry if registry is not None else REGISTRY
        self.controller_instance = (
            controller_instance
            if controller_instance is not None
            else os.environ.get("CONTROLLER_INSTANCE", "")
        )
        base_labels = [C.LABEL_VARIANT_NAME, C.LABEL_NAMESPACE]
        extra = [C.LABEL_CONTROLLER_INSTANCE] if self.controller_instance else []
        self.replica_scaling_total = Counter(
            C.WVA_REPLICA_SCALING_TOTAL,
            "Total number of replica scaling operations",
            base_labels + [C.LABEL_DIRECTION, C.LABEL_REASON] + extra,
            registry=self.registry,
        )
        gauge_labels = base_labels + [C.LABEL_ACCELERATOR_TYPE] + extra
        self.desired_replicas = Gauge(
            C.WVA_DESIRED_REPLICAS,
            "Desired number of replicas",
            gauge_labels,
            registry=self.registry,
        )
        self.current_replicas = Gauge(
            C.WVA_CURRENT_REPLICAS,
            "Current number of replicas",
            gauge_labels,
            registry=self.registry,
        )
        self.desired_ratio = Gauge(
            C.WVA_DESIRED_RATIO,
            "Ratio of desired to current replicas",
            gauge_labels,
            registry=self.registry,
        )

    def _with_instance(self, labels: dict) -> dict:
        if self.controller_instance:
            labels[C.LABEL_CONTROLLER_INSTANCE] = self.controller_instance
        return labels

    def emit_replica_scaling_metrics(
        self, variant_name: str, namespace: str, direction: str, reason: str
    ) -> None:
        labels = self._with_instance(
            {
                C.LABEL_VARIANT_NAME: variant_name,
                C.LABEL_NAMESPACE: namespace,
                C.LABEL_DIRECTION: direction,
                C.LABEL_REASON: reason,
            }
        )
        self.replica_scaling_total.labels(**labels).inc()

    def emit_replica_metrics(
        self,
        variant_name: str,
        namespace: str,
        current: int,
        desired: int,
        accelerator_type: str,
    ) -> None:
        labels = self._with_instance(
            {
                C.LABEL_VARIANT_NAME: variant_name,
                C.LABEL_NAMESPACE: namespace,
                C.LABEL_ACCELERATOR_TYPE: accelerator_type,
            }
        )
        self.current_replicas.labels(**labels).set(current)
        self.desired_replicas.labels(**labels).set(desired)
        if current == 0:
            self.desired_ratio.labels(**labels).set(desired)
        else:
            self.desired_ratio.labels(**labels).set(desired / current)


def init_metrics(
    registry: Optional[CollectorRegistry] = None,
    controller_instance: Optional[str] = None,
) -> MetricsEmitter:
    """Initialize (or return) the process-wide emitter."""
    global _emitter
    with _lock:
        if _emitter is None or registry is not None:
            _emitter = MetricsEmitter(registry, controller_instance)
        return _emitter
