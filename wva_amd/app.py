"""Process wiring — the cmd/main.go analog.

Builds the full controller stack: config, datastore, metrics source,
collector, engines, reconcilers, manager. Parity with reference
cmd/main.go:83-520 (flags+config load, ConfigMap bootstrap before
runnables, Prometheus validation, engine runnables, reconciler setup,
metric registration, manager start).

Used three ways:
  * tests/emulator: build_app(cluster, source=SimMetricsSource(sim))
  * bench.py: same, with simulated time driven manually
  * `python -m wva_amd.app`: real-cluster mode is not wired in this
    environment (no Kubernetes API available) — the FakeCluster is the
    only backend; a REST-backed client can implement the same surface.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from prometheus_client import CollectorRegistry

from .actuator.actuator import Actuator
from .actuator.direct import DirectActuator
from .analyzers.capacity_store import CapacityKnowledgeStore
from .collector import registration as reg
from .collector.pod_va_mapper import PodVAMapper
from .collector.prometheus_source import PrometheusSource
from .collector.registry import SourceRegistry
from .collector.replica_metrics import ReplicaMetricsCollector
from .collector.source import MetricsSource
from .config.config import Config
from .controllers.configmap import ConfigMapReconciler
from .controllers.inferencepool import InferencePoolReconciler
from .controllers.predicates import (
    configmap_predicate,
    deployment_predicate,
    inferencepool_predicate,
    variant_autoscaling_predicate,
)
from .controllers.variantautoscaling import VariantAutoscalingReconciler
from .datastore.datastore import Datastore
from .engines.common import DecisionCache, DecisionTrigger
from .engines.saturation import SaturationEngine
from .engines.scalefromzero import ScaleFromZeroEngine
from .kube.fake import FakeCluster, WatchEvent
from .kube.indexers import VAIndex
from .metrics.metrics import MetricsEmitter
from .pipeline.enforcer import Enforcer
from .pipeline.greedy_saturation import GreedyBySaturation
from .pipeline.inventory import TypeInventory
from .pipeline.limiter import DefaultLimiter
from .discovery.gpu_operator import K8sGpuOperatorDiscovery
from .runtime.manager import Manager
from .utils.logging import get_logger

log = get_logger("app")


@dataclass
class App:
    cluster: FakeCluster
    config: Config
    manager: Manager
    datastore: Datastore
    source_registry: SourceRegistry
    collector: ReplicaMetricsCollector
    saturation_engine: SaturationEngine
    scale_from_zero_engine: ScaleFromZeroEngine
    va_reconciler: VariantAutoscalingReconciler
    configmap_reconciler: ConfigMapReconciler
    inferencepool_reconciler: InferencePoolReconciler
    decision_cache: DecisionCache
    decision_trigger: DecisionTrigger
    emitter: MetricsEmitter
    capacity_store: CapacityKnowledgeStore
    va_index: VAIndex
    probe_server: Optional["ProbeServer"] = None
    metrics_server: Optional["ProbeServer"] = None

    def start(self) -> None:
        self.manager.start()
        if self.probe_server is not None:
            self.probe_server.start()
        if self.metrics_server is not None:
            self.metrics_server.start()

    def stop(self) -> None:
        if self.probe_server is not None:
            self.probe_server.stop()
        if self.metrics_server is not None:
            self.metrics_server.stop()
        self.manager.stop()


def build_app(
    cluster: FakeCluster,
    config: Config,
    source: Optional[MetricsSource] = None,
    metrics_registry: Optional[CollectorRegistry] = None,
    engine_interval_seconds: float = 30.0,
    sfz_interval_seconds: float = 0.1,
    scrape_fetch=None,
    start_engines: bool = True,
    serve_http: bool = False,
    measured_profiles_dir: Optional[str] = None,
) -> App:
    # Metrics source: explicit (sim) or Prometheus from config
    source_registry = SourceRegistry()
    if source is None:
        source = PrometheusSource(
            config.prometheus.base_url,
            cache_ttl_seconds=config.cache.ttl_seconds,
            bearer_token=config.prometheus.bearer_token,
            token_path=config.prometheus.token_path,
            verify_tls=not config.prometheus.insecure_skip_verify,
            ca_cert_path=config.prometheus.ca_cert_path,
            client_cert_path=config.prometheus.client_cert_path,
            client_key_path=config.prometheus.client_key_path,
        )
    source_registry.register(source)
    reg.register_saturation_queries(source_registry)
    reg.register_scale_to_zero_queries(source_registry)
    reg.register_arrival_rate_query(source_registry)
    reg.register_latency_queries(source_registry)

    epp_secret_name, epp_secret_key = config.epp_metrics_reader_secret()
    datastore = Datastore(
        cluster,
        epp_bearer_token=config.epp_metric_reader_bearer_token(),
        epp_metrics_reader_secret_name=epp_secret_name,
        epp_metrics_reader_secret_key=epp_secret_key,
        scrape_fetch=scrape_fetch,
        source_registry=source_registry,
    )
    emitter = MetricsEmitter(registry=metrics_registry)
    decision_cache = DecisionCache()
    decision_trigger = DecisionTrigger()

    collector = ReplicaMetricsCollector(
        source, PodVAMapper(cluster), freshness=config.freshness
    )
    actuator = Actuator(cluster, emitter)
    enforcer = Enforcer(
        lambda model_id, namespace, retention: reg.collect_model_request_count(
            source, model_id, namespace, retention
        )
    )
    discovery = K8sGpuOperatorDiscovery(cluster)
    inventory = TypeInventory("gpu", discovery)
    limiter = DefaultLimiter("gpu-limiter", inventory, GreedyBySaturation())
    capacity_store = CapacityKnowledgeStore()
    if measured_profiles_dir:
        # seed the MI355X calibration registry (profiles/*.json): new
        # variants get hardware-measured k1 (incl. fp8 2x) before any
        # live cache_config_info arrives
        from .analyzers.capacity_store import load_calibration_dir

        n = load_calibration_dir(capacity_store, measured_profiles_dir)
        log.info("loaded %d measured capacity profiles from %s",
                 n, measured_profiles_dir)

    saturation_engine = SaturationEngine(
        cluster=cluster,
        config=config,
        collector=collector,
        enforcer=enforcer,
        actuator=actuator,
        decision_cache=decision_cache,
        decision_trigger=decision_trigger,
        limiter=limiter,
        capacity_store=capacity_store,
        interval_seconds=engine_interval_seconds,
    )
    scale_from_zero_engine = ScaleFromZeroEngine(
        cluster=cluster,
        config=config,
        datastore=datastore,
        direct_actuator=DirectActuator(cluster),
        decision_cache=decision_cache,
        decision_trigger=decision_trigger,
        interval_seconds=sfz_interval_seconds,
    )

    va_reconciler = VariantAutoscalingReconciler(cluster, datastore, decision_cache)
    configmap_reconciler = ConfigMapReconciler(cluster, config, datastore)
    inferencepool_reconciler = InferencePoolReconciler(cluster, datastore)

    manager = Manager(cluster, config, decision_trigger)
    manager.register_reconciler(
        ["VariantAutoscaling", "Deployment"],
        _va_or_deployment_predicate(cluster),
        va_reconciler.reconcile,
        map_func=_map_to_va(cluster),
        is_va_reconciler=True,
    )
    manager.register_reconciler(
        ["ConfigMap"], configmap_predicate(), configmap_reconciler.reconcile
    )
    manager.register_reconciler(
        ["InferencePool"],
        inferencepool_predicate(),
        inferencepool_reconciler.reconcile,
    )

    # ServiceMonitor deletion → observability warning (controller:330-367):
    # without the monitor, Prometheus stops scraping vLLM pods and the
    # saturation metrics go stale.
    def _servicemonitor_warning(ns: str, name: str) -> None:
        if cluster.try_get("ServiceMonitor", ns, name) is None:
            log.warning(
                "ServiceMonitor %s/%s deleted — vLLM metrics scraping will "
                "stop and saturation decisions will degrade to the safety net",
                ns, name,
            )

    from .kube.fake import DELETED as _DELETED

    manager.register_reconciler(
        ["ServiceMonitor"],
        lambda e: e.kind == "ServiceMonitor" and e.type == _DELETED,
        _servicemonitor_warning,
    )
    if start_engines:
        manager.add_runnable(saturation_engine)
        manager.add_runnable(scale_from_zero_engine)

    # ConfigMap bootstrap before runnables (cmd/main.go:322-336)
    configmap_reconciler.bootstrap_initial_configmaps()

    # capacity-store persistence (checkpoint/resume improvement): restore
    # learned capacity records from the wva-capacity-store ConfigMap and
    # write them back periodically from the engine tick
    from .analyzers.capacity_store import CapacityStorePersistence
    from .controllers.configmap import controller_namespace

    persistence = CapacityStorePersistence(
        cluster, capacity_store, controller_namespace(),
        analyzer=saturation_engine.v2_analyzer,
    )
    restored = persistence.restore()
    if restored:
        log.info("restored %d capacity records from %s/%s",
                 restored, controller_namespace(), "wva-capacity-store")
    saturation_engine.capacity_persistence = persistence

    # HTTP surfaces (cmd/main.go:266-287,482-498): probe address serves
    # /healthz + /readyz (readyz gated on ConfigMap bootstrap); metrics
    # address serves /metrics from the emitter's registry.
    probe_server = metrics_server = None
    if serve_http:
        from .runtime.http import ProbeServer

        probe_addr = config.infra.health_probe_bind_address
        metrics_addr = config.infra.metrics_bind_address
        if probe_addr and probe_addr != "0":
            probe_server = ProbeServer(
                probe_addr,
                healthz=manager.healthz,
                readyz=manager.readyz,
                serve_metrics=False,
            )
        if metrics_addr and metrics_addr != "0":
            watcher = None
            infra = config.infra
            if infra.metrics_cert_path and infra.metrics_key_path:
                from .runtime.http import CertWatcher

                watcher = CertWatcher(
                    infra.metrics_cert_path, infra.metrics_key_path
                )
            metrics_server = ProbeServer(
                metrics_addr,
                healthz=manager.healthz,
                readyz=manager.readyz,
                registry=emitter.registry,
                cert_watcher=watcher,
            )

    return App(
        cluster=cluster,
        config=config,
        manager=manager,
        datastore=datastore,
        source_registry=source_registry,
        collector=collector,
        saturation_engine=saturation_engine,
        scale_from_zero_engine=scale_from_zero_engine,
        va_reconciler=va_reconciler,
        configmap_reconciler=configmap_reconciler,
        inferencepool_reconciler=inferencepool_reconciler,
        decision_cache=decision_cache,
        decision_trigger=decision_trigger,
        emitter=emitter,
        capacity_store=capacity_store,
        va_index=VAIndex(cluster),
        probe_server=probe_server,
        metrics_server=metrics_server,
    )


def _va_or_deployment_predicate(cluster: FakeCluster):
    va_pred = variant_autoscaling_predicate(cluster)
    dep_pred = deployment_predicate()

    def pred(event: WatchEvent) -> bool:
        if event.kind == "VariantAutoscaling":
            return va_pred(event)
        return dep_pred(event)

    return pred


def _map_to_va(cluster: FakeCluster):
    index = VAIndex(cluster)

    def map_func(event: WatchEvent):
        if event.kind == "VariantAutoscaling":
            return (event.obj.metadata.namespace, event.obj.metadata.name)
        # Deployment event → indexed VA lookup (controller:258-288)
        try:
            va = index.find_va_for_deployment(
                event.obj.metadata.namespace, event.obj.metadata.name
            )
        except ValueError:
            return None
        if va is None:
            return None
        return (va.namespace, va.name)

    return map_func
