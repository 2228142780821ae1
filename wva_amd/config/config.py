"""Unified, thread-safe runtime configuration.

Parity: reference internal/config/config.go:15-631 — a single RWMutex-guarded
struct holding infrastructure settings, TLS/Prometheus connection config,
feature flags, and namespace-aware saturation / scale-to-zero config maps
with namespace-local > global resolution. All access is via getters/updaters
holding the lock.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Dict

from .saturation import SaturationScalingConfig
from .scale_to_zero import (
    ScaleToZeroConfigData,
    is_scale_to_zero_enabled,
    scale_to_zero_retention_seconds,
)


@dataclass
class PrometheusConfig:
    """Prometheus connection + TLS config (reference internal/config/prometheus.go)."""

    base_url: str = ""
    bearer_token: str = ""
    token_path: str = ""
    insecure_skip_verify: bool = False
    ca_cert_path: str = ""
    client_cert_path: str = ""
    client_key_path: str = ""
    server_name: str = ""


@dataclass
class CacheConfig:
    """Metrics TTL-cache config (prometheus_source 30s TTL default)."""

    ttl_seconds: float = 30.0
    cleanup_interval_seconds: float = 60.0
    fetch_interval_seconds: float = 0.0


@dataclass
class FreshnessThresholds:
    """Freshness ladder (reference internal/config/prometheus.go FreshnessThresholds)."""

    fresh_seconds: float = 60.0
    stale_seconds: float = 120.0
    unavailable_seconds: float = 300.0

    def determine_status(self, age_seconds: float) -> str:
        if age_seconds <= self.fresh_seconds:
            return "fresh"
        if age_seconds <= self.stale_seconds:
            return "stale"
        return "unavailable"


@dataclass
class InfrastructureConfig:
    metrics_bind_address: str = "0"
    health_probe_bind_address: str = ":8081"
    enable_leader_election: bool = False
    leader_election_id: str = "72dd1cf1.llm-d.ai"
    lease_duration_seconds: float = 60.0
    renew_deadline_seconds: float = 50.0
    retry_period_seconds: float = 10.0
    rest_client_timeout_seconds: float = 60.0
    secure_metrics: bool = True
    enable_http2: bool = False
    metrics_cert_path: str = ""   # TLS cert for the metrics listener
    metrics_key_path: str = ""    # (cert-manager mounts; hot-reloaded)
    watch_namespace: str = ""
    logger_verbosity: int = 0
    optimization_interval_seconds: float = 60.0


class Config:
    """Process-wide configuration registry. Thread-safe."""

    def __init__(self) -> None:
        self._lock = threading.RLock()
        self.infra = InfrastructureConfig()
        self.prometheus = PrometheusConfig()
        self.cache = CacheConfig()
        self.freshness = FreshnessThresholds()
        self._scale_to_zero_enabled = False
        self._limited_mode_enabled = False
        self._scale_from_zero_max_concurrency = 10
        self._epp_metric_reader_bearer_token = ""
        self._epp_metrics_reader_secret_name = ""
        self._epp_metrics_reader_secret_key = "token"
        # global saturation config + per-namespace overrides
        self._saturation_global = SaturationScalingConfig()
        self._saturation_by_ns: Dict[str, SaturationScalingConfig] = {}
        # global scale-to-zero config + per-namespace overrides
        self._stz_global: ScaleToZeroConfigData = {}
        self._stz_by_ns: Dict[str, ScaleToZeroConfigData] = {}
        self._bootstrap_complete = False
        # Inferno system config (service classes + accelerators + model
        # perf data) from the wva-{service-class,accelerator,model-perf}
        # ConfigMaps; version bumps on every update so the engine can
        # rebuild its analyzer lazily on live reload
        self._inferno_service_classes: list = []
        self._inferno_accelerators: list = []
        self._inferno_perf: list = []
        self._inferno_version = 0

    # --- feature flags ---

    def scale_to_zero_enabled(self) -> bool:
        with self._lock:
            return self._scale_to_zero_enabled

    def set_scale_to_zero_enabled(self, v: bool) -> None:
        with self._lock:
            self._scale_to_zero_enabled = v

    def limited_mode_enabled(self) -> bool:
        with self._lock:
            return self._limited_mode_enabled

    def set_limited_mode_enabled(self, v: bool) -> None:
        with self._lock:
            self._limited_mode_enabled = v

    def scale_from_zero_max_concurrency(self) -> int:
        with self._lock:
            return self._scale_from_zero_max_concurrency

    def set_scale_from_zero_max_concurrency(self, v: int) -> None:
        with self._lock:
            self._scale_from_zero_max_concurrency = max(1, int(v))

    def epp_metric_reader_bearer_token(self) -> str:
        with self._lock:
            return self._epp_metric_reader_bearer_token

    def set_epp_metric_reader_bearer_token(self, v: str) -> None:
        with self._lock:
            self._epp_metric_reader_bearer_token = v

    def epp_metrics_reader_secret(self) -> tuple:
        """(secret_name, key) for the Secret-sourced EPP bearer token
        (reference pod_scraping_source.go:300-331)."""
        with self._lock:
            return (
                self._epp_metrics_reader_secret_name,
                self._epp_metrics_reader_secret_key,
            )

    def set_epp_metrics_reader_secret(self, name: str, key: str = "token") -> None:
        with self._lock:
            self._epp_metrics_reader_secret_name = name
            self._epp_metrics_reader_secret_key = key or "token"

    # --- saturation config (global vs namespace-local) ---

    def update_saturation_config(self, cfg: SaturationScalingConfig) -> None:
        with self._lock:
            self._saturation_global = cfg

    def update_saturation_config_for_namespace(
        self, namespace: str, cfg: SaturationScalingConfig
    ) -> None:
        with self._lock:
            self._saturation_by_ns[namespace] = cfg

    def remove_saturation_config_for_namespace(self, namespace: str) -> None:
        with self._lock:
            self._saturation_by_ns.pop(namespace, None)

    def saturation_config_for_namespace(
        self, namespace: str
    ) -> SaturationScalingConfig:
        """Namespace-local config wins over global (config.go:360)."""
        with self._lock:
            return self._saturation_by_ns.get(namespace, self._saturation_global)

    def saturation_config(self) -> SaturationScalingConfig:
        with self._lock:
            return self._saturation_global

    # --- scale-to-zero config (global vs namespace-local) ---

    def update_scale_to_zero_config(self, data: ScaleToZeroConfigData) -> None:
        with self._lock:
            self._stz_global = data

    def update_scale_to_zero_config_for_namespace(
        self, namespace: str, data: ScaleToZeroConfigData
    ) -> None:
        with self._lock:
            self._stz_by_ns[namespace] = data

    def remove_scale_to_zero_config_for_namespace(self, namespace: str) -> None:
        with self._lock:
            self._stz_by_ns.pop(namespace, None)

    def scale_to_zero_config_for_namespace(
        self, namespace: str
    ) -> ScaleToZeroConfigData:
        with self._lock:
            return self._stz_by_ns.get(namespace, self._stz_global)

    def remove_namespace_config(self, namespace: str) -> None:
        with self._lock:
            self._saturation_by_ns.pop(namespace, None)
            self._stz_by_ns.pop(namespace, None)

    # --- resolved scale-to-zero queries ---

    def is_scale_to_zero_enabled_for(self, model_id: str, namespace: str) -> bool:
        data = self.scale_to_zero_config_for_namespace(namespace)
        if data:
            return is_scale_to_zero_enabled(data, model_id)
        return self.scale_to_zero_enabled()

    def scale_to_zero_retention_seconds_for(
        self, model_id: str, namespace: str
    ) -> float:
        return scale_to_zero_retention_seconds(
            self.scale_to_zero_config_for_namespace(namespace), model_id
        )

    def min_num_replicas_for(self, model_id: str, namespace: str) -> int:
        if self.is_scale_to_zero_enabled_for(model_id, namespace):
            return 0
        return 1

    # --- bootstrap gating (readyz depends on this, cmd/main.go:486-498) ---

    # --- Inferno system config (live-reloadable SLO analyzer inputs) ---

    def update_inferno_service_classes(self, specs: list) -> None:
        with self._lock:
            self._inferno_service_classes = list(specs)
            self._inferno_version += 1

    def update_inferno_accelerators(self, specs: list) -> None:
        with self._lock:
            self._inferno_accelerators = list(specs)
            self._inferno_version += 1

    def update_inferno_perf(self, specs: list) -> None:
        with self._lock:
            self._inferno_perf = list(specs)
            self._inferno_version += 1

    def inferno_config_version(self) -> int:
        with self._lock:
            return self._inferno_version

    def inferno_system_data(self):
        """Build a SystemData from the ConfigMap-fed pieces; None until
        accelerators, perf data AND service classes have all arrived
        (the analyzer cannot size replicas without SLO targets and
        measured service parms)."""
        from ..inferno.types import SystemData

        with self._lock:
            if not (
                self._inferno_accelerators
                and self._inferno_perf
                and self._inferno_service_classes
            ):
                return None
            return SystemData(
                accelerators=list(self._inferno_accelerators),
                models=list(self._inferno_perf),
                service_classes=list(self._inferno_service_classes),
            )

    def mark_bootstrap_complete(self) -> None:
        with self._lock:
            self._bootstrap_complete = True

    def is_bootstrap_complete(self) -> bool:
        with self._lock:
            return self._bootstrap_complete
