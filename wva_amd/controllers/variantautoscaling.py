"""VariantAutoscaling reconciler.

Parity: reference internal/controller/variantautoscaling_controller.go
:90-319. Per reconcile:
  1. fetch VA (deep copy); deletion → untrack namespace, done
  2. track namespace in the datastore
  3. resolve scaleTargetRef Deployment with backoff → TargetResolved
     condition (not found: persist condition, no requeue)
  4. consume the engine decision from the DecisionCache → write
     status.desiredOptimizedAlloc + MetricsAvailable condition
  5. persist via the status subresource, writing the FULL nested
     desiredOptimizedAlloc object (reference #731: partial merge patches
     are rejected by CRD validation; FakeCluster.update_status replaces
     the whole status, preserving those semantics)

The reconciler is the ONLY component that persists VA status through the
API; the engine feeds it via DecisionCache + DecisionTrigger.
"""
from __future__ import annotations

from typing import Optional

from ..api import conditions as cond
from ..api.types import (
    REASON_METRICS_FOUND,
    REASON_METRICS_MISSING,
    REASON_TARGET_FOUND,
    REASON_TARGET_NOT_FOUND,
    TYPE_METRICS_AVAILABLE,
    TYPE_TARGET_RESOLVED,
    OptimizedAlloc,
    VariantAutoscaling,
)
from ..datastore.datastore import Datastore
from ..engines.common import DecisionCache
from ..kube.fake import ConflictError, FakeCluster, NotFoundError
from ..utils.backoff import retry_with_backoff
from ..utils.logging import get_logger
from ..utils.variant import matches_controller_instance, namespace_excluded

log = get_logger("controllers.va")


class VariantAutoscalingReconciler:
    def __init__(
        self,
        cluster: FakeCluster,
        datastore: Datastore,
        decision_cache: DecisionCache,
    ):
        self.cluster = cluster
        self.datastore = datastore
        self.decision_cache = decision_cache

    def reconcile(self, namespace: str, name: str) -> None:
        va: Optional[VariantAutoscaling] = self.cluster.try_get(
            "VariantAutoscaling", namespace, name
        )
        if va is None:
            # Deleted: stop tracking the namespace if it has no more VAs
            remaining = self.cluster.list("VariantAutoscaling", namespace=namespace)
            if not remaining:
                self.datastore.namespace_untrack(namespace)
            self.decision_cache.delete(namespace, name)
            return

        if va.metadata.deletion_timestamp is not None:
            return
        if not matches_controller_instance(va):
            return
        if namespace_excluded(self.cluster, namespace):
            return

        self.datastore.namespace_track(namespace)

        # Resolve scale target
        def fetch():
            return self.cluster.get(
                va.get_scale_target_kind(), namespace, va.get_scale_target_name()
            )

        try:
            retry_with_backoff(
                fetch, retry_on=NotFoundError, max_attempts=2, initial_delay=0.01
            )
            cond.set_condition(
                va,
                TYPE_TARGET_RESOLVED,
                "True",
                REASON_TARGET_FOUND,
                f"Scale target {va.get_scale_target_name()} resolved",
            )
        except NotFoundError:
            cond.set_condition(
                va,
                TYPE_TARGET_RESOLVED,
                "False",
                REASON_TARGET_NOT_FOUND,
                f"Scale target {va.get_scale_target_name()} not found",
            )
            self.cluster.record_event(
                va, "Warning", REASON_TARGET_NOT_FOUND,
                f"scale target Deployment {va.get_scale_target_name()} "
                "not found",
            )
            self._patch_status(va)
            return  # no requeue — Deployment create watch will retrigger

        # Consume engine decision
        decision = self.decision_cache.get(namespace, name)
        if decision is not None:
            if decision.accelerator_name or decision.target_replicas:
                va.status.desired_optimized_alloc = OptimizedAlloc(
                    last_run_time=decision.last_run_time,
                    accelerator=decision.accelerator_name,
                    num_replicas=decision.target_replicas,
                )
            if decision.metrics_available:
                cond.set_condition(
                    va,
                    TYPE_METRICS_AVAILABLE,
                    "True",
                    decision.metrics_reason or REASON_METRICS_FOUND,
                    decision.metrics_message,
                )
            else:
                cond.set_condition(
                    va,
                    TYPE_METRICS_AVAILABLE,
                    "False",
                    decision.metrics_reason or REASON_METRICS_MISSING,
                    decision.metrics_message,
                )
            if decision.optimization_ready_reason:
                cond.set_condition(
                    va,
                    "OptimizationReady",
                    "True",
                    decision.optimization_ready_reason,
                    decision.optimization_ready_message,
                )

        self._patch_status(va)

    def _patch_status(self, va: VariantAutoscaling) -> None:
        try:
            self.cluster.update_status(va)
        except NotFoundError:
            log.debug("VA %s deleted during reconcile", va.full_name())
        except ConflictError:
            # a competing writer bumped the VA between our read and this
            # write — controller-runtime would requeue the whole
            # reconcile; one retry against a fresh read is the same
            # convergence without re-running target resolution
            fresh = self.cluster.try_get(
                "VariantAutoscaling", va.namespace, va.name
            )
            if fresh is None:
                return
            fresh.status = va.status
            try:
                self.cluster.update_status(fresh)
            except (NotFoundError, ConflictError) as e:
                # the next watch event for the competing write retriggers
                log.debug("status retry for %s dropped: %s",
                          va.full_name(), e)
