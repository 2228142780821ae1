"""Scale-from-zero engine — the 100 ms reactive loop.

Parity: reference internal/engines/scalefromzero/engine.go:73-358. Per
tick: list inactive VAs (deployment replicas == 0), process each under a
concurrency bound (SCALE_FROM_ZERO_ENGINE_MAX_CONCURRENCY, default 10);
per VA: resolve the target deployment's pod-template labels, find the
owning EndpointPool, read the EPP pod-scrape source's `all_metrics`, and
when `inference_extension_flow_control_queue_size{target_model_name ==
modelID} > 0`, scale 0→1 through the DirectActuator, seed the
DecisionCache, set status + the ScaleFromZeroMode condition and push the
DecisionTrigger.
"""
from __future__ import annotations

import concurrent.futures

from ..actuator.direct import DirectActuator
from ..analyzers.interfaces import ACTION_SCALE_UP, VariantDecision
from ..api import conditions as cond
from ..api.types import OptimizedAlloc, VariantAutoscaling, utcnow
from ..collector.pod_scraping_source import ALL_METRICS_QUERY
from ..config.config import Config
from ..constants import ACCELERATOR_LABEL_KEY, SCHEDULER_FLOW_CONTROL_QUEUE_SIZE
from ..datastore.datastore import Datastore
from ..kube.fake import FakeCluster
from ..runtime.executor import PollingExecutor
from ..utils.logging import get_logger
from ..utils.variant import inactive_variant_autoscalings
from .common import DecisionCache, DecisionTrigger

log = get_logger("engines.scalefromzero")

DEFAULT_SFZ_INTERVAL_SECONDS = 0.1


class ScaleFromZeroEngine:
    def __init__(
        self,
        cluster: FakeCluster,
        config: Config,
        datastore: Datastore,
        direct_actuator: DirectActuator,
        decision_cache: DecisionCache,
        decision_trigger: DecisionTrigger,
        interval_seconds: float = DEFAULT_SFZ_INTERVAL_SECONDS,
    ):
        self.cluster = cluster
        self.config = config
        self.datastore = datastore
        self.direct_actuator = direct_actuator
        self.decision_cache = decision_cache
        self.decision_trigger = decision_trigger
        self.executor = PollingExecutor(
            interval_seconds, self.optimize, name="scale-from-zero-engine"
        )

    def start(self) -> None:
        self.executor.start()

    def stop(self) -> None:
        self.executor.stop()

    def optimize(self) -> None:
        watch_ns = self.config.infra.watch_namespace or None
        inactive = inactive_variant_autoscalings(self.cluster, watch_ns)
        if not inactive:
            return
        max_conc = self.config.scale_from_zero_max_concurrency()
        if len(inactive) == 1 or max_conc <= 1:
            for va in inactive:
                self._process_inactive_variant(va)
            return
        with concurrent.futures.ThreadPoolExecutor(
            max_workers=min(max_conc, len(inactive))
        ) as pool:
            futures = [
                pool.submit(self._process_inactive_variant, va) for va in inactive
            ]
            for fut in futures:
                try:
                    fut.result()
                except Exception as e:  # noqa: BLE001
                    log.error("scale-from-zero processing failed: %s", e)

    # --- per-VA processing ---

    def _process_inactive_variant(self, va: VariantAutoscaling) -> None:
        deploy = self.cluster.try_get(
            va.get_scale_target_kind(), va.namespace, va.get_scale_target_name()
        )
        if deploy is None:
            log.debug("target %s not found for VA %s", va.get_scale_target_name(),
                      va.full_name())
            return

        pool = self.datastore.pool_get_from_labels(
            va.namespace, deploy.template.labels
        )
        if pool is None:
            log.debug("no EndpointPool for VA %s", va.full_name())
            return

        if not self._has_pending_requests(pool, va.spec.model_id):
            return

        log.info(
            "pending requests detected for idle model %s — scaling %s 0→1",
            va.spec.model_id,
            va.full_name(),
        )
        try:
            self.direct_actuator.scale_target_object(
                va.get_scale_target_kind(), va.namespace,
                va.get_scale_target_name(), 1,
            )
        except Exception as e:  # noqa: BLE001
            log.error("scale 0→1 failed for %s: %s", va.full_name(), e)
            return

        accelerator = va.status.desired_optimized_alloc.accelerator or (
            va.metadata.labels.get(ACCELERATOR_LABEL_KEY, "")
        )
        decision = VariantDecision(
            variant_name=va.name,
            namespace=va.namespace,
            model_id=va.spec.model_id,
            accelerator_name=accelerator,
            action=ACTION_SCALE_UP,
            current_replicas=0,
            target_replicas=1,
            last_run_time=utcnow(),
            reason="scale-from-zero: pending requests in scheduler queue",
            metrics_available=True,
            metrics_reason="MetricsFound",
            metrics_message="scale-from-zero triggered by EPP flow-control queue",
        )
        self.decision_cache.set(va.namespace, va.name, decision)

        # Update VA status directly (scale-from-zero is reactive and the
        # reconciler persists through the cache + trigger as usual)
        update_va = self.cluster.try_get(
            "VariantAutoscaling", va.namespace, va.name
        )
        if update_va is not None:
            update_va.status.desired_optimized_alloc = OptimizedAlloc(
                last_run_time=utcnow(),
                accelerator=accelerator,
                num_replicas=1,
            )
            cond.set_condition(
                update_va,
                "ScaleFromZeroMode",
                "True",
                "PendingRequests",
                "scaled from zero due to pending requests",
            )
            try:
                self.cluster.update_status(update_va)
            except Exception as e:  # noqa: BLE001
                log.error("status update failed for %s: %s", va.full_name(), e)
        self.decision_trigger.push(va.namespace, va.name)

    def _has_pending_requests(self, pool, model_id: str) -> bool:
        source = self.datastore.pool_source(pool.namespace, pool.name)
        if source is None:
            return False
        from ..collector.source import RefreshSpec

        try:
            results = source.refresh(RefreshSpec(queries=[ALL_METRICS_QUERY]))
        except Exception as e:  # noqa: BLE001
            log.debug("EPP scrape failed for pool %s: %s", pool.name, e)
            return False
        result = results.get(ALL_METRICS_QUERY)
        if result is None or result.has_error():
            return False
        for v in result.values:
            if v.labels.get("__name__") != SCHEDULER_FLOW_CONTROL_QUEUE_SIZE:
                continue
            target = v.labels.get("target_model_name", "")
            name = v.labels.get("model_name", "")
            if (target == model_id or (not target and name == model_id)) and v.value > 0:
                return True
        return False
