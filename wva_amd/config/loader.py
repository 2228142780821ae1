"""Config loader with precedence flags > env > config file > defaults.

Parity: reference internal/config/loader.go:13-219 (viper). Same keys and
defaults (METRICS_BIND_ADDRESS "0", GLOBAL_OPT_INTERVAL "60s",
PROMETHEUS_BASE_URL required, WVA_SCALE_TO_ZERO / WVA_LIMITED_MODE feature
flags, SCALE_FROM_ZERO_ENGINE_MAX_CONCURRENCY 10, Prometheus TLS + cache
keys). A config file is YAML with the same UPPER_SNAKE keys.
"""
from __future__ import annotations

import os
from typing import Any, Dict, Optional

import yaml

from .config import Config
from .scale_to_zero import parse_go_duration
from .validation import ConfigLoadError, validate_config

_DEFAULTS: Dict[str, Any] = {
    "METRICS_BIND_ADDRESS": "0",
    "HEALTH_PROBE_BIND_ADDRESS": ":8081",
    "LEADER_ELECT": False,
    "LEADER_ELECTION_ID": "72dd1cf1.llm-d.ai",
    "LEADER_ELECTION_LEASE_DURATION": "60s",
    "LEADER_ELECTION_RENEW_DEADLINE": "50s",
    "LEADER_ELECTION_RETRY_PERIOD": "10s",
    "REST_CLIENT_TIMEOUT": "60s",
    "METRICS_SECURE": True,
    "METRICS_CERT_PATH": "",
    "METRICS_KEY_PATH": "",
    "ENABLE_HTTP2": False,
    "WATCH_NAMESPACE": "",
    "V": 0,
    "WVA_SCALE_TO_ZERO": False,
    "WVA_LIMITED_MODE": False,
    "SCALE_FROM_ZERO_ENGINE_MAX_CONCURRENCY": 10,
    "EPP_METRIC_READER_BEARER_TOKEN": "",
    "EPP_METRICS_READER_SECRET_NAME": "",
    "EPP_METRICS_READER_SECRET_KEY": "token",
    "GLOBAL_OPT_INTERVAL": "60s",
    "PROMETHEUS_BASE_URL": "",
    "PROMETHEUS_BEARER_TOKEN": "",
    "PROMETHEUS_TOKEN_PATH": "",
    "PROMETHEUS_TLS_INSECURE_SKIP_VERIFY": False,
    "PROMETHEUS_CA_CERT_PATH": "",
    "PROMETHEUS_CLIENT_CERT_PATH": "",
    "PROMETHEUS_CLIENT_KEY_PATH": "",
    "PROMETHEUS_SERVER_NAME": "",
    "PROMETHEUS_METRICS_CACHE_TTL": "30s",
    "PROMETHEUS_METRICS_CACHE_CLEANUP_INTERVAL": "60s",
    "PROMETHEUS_METRICS_CACHE_FETCH_INTERVAL": "0s",
    "PROMETHEUS_METRICS_CACHE_FRESH_THRESHOLD": "1m",
    "PROMETHEUS_METRICS_CACHE_STALE_THRESHOLD": "2m",
    "PROMETHEUS_METRICS_CACHE_UNAVAILABLE_THRESHOLD": "5m",
}


def _coerce_bool(v: Any) -> bool:
    if isinstance(v, bool):
        return v
    return str(v).strip().lower() in ("1", "true", "yes", "on")


def _duration_seconds(v: Any) -> float:
    if isinstance(v, (int, float)):
        return float(v)
    return parse_go_duration(str(v))


def load_config(
    flags: Optional[Dict[str, Any]] = None,
    config_path: Optional[str] = None,
    env: Optional[Dict[str, str]] = None,
    require_prometheus: bool = True,
) -> Config:
    """Merge defaults < config file < env < flags, validate, build a Config."""
    env = dict(os.environ) if env is None else env
    merged: Dict[str, Any] = dict(_DEFAULTS)

    if config_path:
        try:
            with open(config_path, "r") as f:
                file_cfg = yaml.safe_load(f) or {}
        except OSError as e:
            raise ConfigLoadError(f"cannot read config file {config_path}: {e}")
        if not isinstance(file_cfg, dict):
            raise ConfigLoadError(f"config file {config_path} must be a mapping")
        merged.update(file_cfg)

    for key in _DEFAULTS:
        if key in env:
            merged[key] = env[key]

    if flags:
        for k, v in flags.items():
            if v is not None:
                merged[k] = v

    cfg = Config()
    infra = cfg.infra
    infra.metrics_bind_address = str(merged["METRICS_BIND_ADDRESS"])
    infra.health_probe_bind_address = str(merged["HEALTH_PROBE_BIND_ADDRESS"])
    infra.enable_leader_election = _coerce_bool(merged["LEADER_ELECT"])
    infra.leader_election_id = str(merged["LEADER_ELECTION_ID"])
    infra.lease_duration_seconds = _duration_seconds(
        merged["LEADER_ELECTION_LEASE_DURATION"]
    )
    infra.renew_deadline_seconds = _duration_seconds(
        merged["LEADER_ELECTION_RENEW_DEADLINE"]
    )
    infra.retry_period_seconds = _duration_seconds(
        merged["LEADER_ELECTION_RETRY_PERIOD"]
    )
    infra.rest_client_timeout_seconds = _duration_seconds(
        merged["REST_CLIENT_TIMEOUT"]
    )
    infra.secure_metrics = _coerce_bool(merged["METRICS_SECURE"])
    infra.metrics_cert_path = str(merged["METRICS_CERT_PATH"])
    infra.metrics_key_path = str(merged["METRICS_KEY_PATH"])
    infra.enable_http2 = _coerce_bool(merged["ENABLE_HTTP2"])
    infra.watch_namespace = str(merged["WATCH_NAMESPACE"])
    infra.logger_verbosity = int(merged["V"])
    infra.optimization_interval_seconds = _duration_seconds(
        merged["GLOBAL_OPT_INTERVAL"]
    )

    cfg.set_scale_to_zero_enabled(_coerce_bool(merged["WVA_SCALE_TO_ZERO"]))
    cfg.set_limited_mode_enabled(_coerce_bool(merged["WVA_LIMITED_MODE"]))
    cfg.set_scale_from_zero_max_concurrency(
        int(merged["SCALE_FROM_ZERO_ENGINE_MAX_CONCURRENCY"])
    )
    cfg.set_epp_metric_reader_bearer_token(
        str(merged["EPP_METRIC_READER_BEARER_TOKEN"])
    )
    cfg.set_epp_metrics_reader_secret(
        str(merged["EPP_METRICS_READER_SECRET_NAME"]),
        str(merged["EPP_METRICS_READER_SECRET_KEY"]),
    )

    prom = cfg.prometheus
    prom.base_url = str(merged["PROMETHEUS_BASE_URL"])
    prom.bearer_token = str(merged["PROMETHEUS_BEARER_TOKEN"])
    prom.token_path = str(merged["PROMETHEUS_TOKEN_PATH"])
    prom.insecure_skip_verify = _coerce_bool(
        merged["PROMETHEUS_TLS_INSECURE_SKIP_VERIFY"]
    )
    prom.ca_cert_path = str(merged["PROMETHEUS_CA_CERT_PATH"])
    prom.client_cert_path = str(merged["PROMETHEUS_CLIENT_CERT_PATH"])
    prom.client_key_path = str(merged["PROMETHEUS_CLIENT_KEY_PATH"])
    prom.server_name = str(merged["PROMETHEUS_SERVER_NAME"])

    cache = cfg.cache
    cache.ttl_seconds = _duration_seconds(merged["PROMETHEUS_METRICS_CACHE_TTL"])
    cache.cleanup_interval_seconds = _duration_seconds(
        merged["PROMETHEUS_METRICS_CACHE_CLEANUP_INTERVAL"]
    )
    cache.fetch_interval_seconds = _duration_seconds(
        merged["PROMETHEUS_METRICS_CACHE_FETCH_INTERVAL"]
    )
    cfg.freshness.fresh_seconds = _duration_seconds(
        merged["PROMETHEUS_METRICS_CACHE_FRESH_THRESHOLD"]
    )
    cfg.freshness.stale_seconds = _duration_seconds(
        merged["PROMETHEUS_METRICS_CACHE_STALE_THRESHOLD"]
    )
    cfg.freshness.unavailable_seconds = _duration_seconds(
        merged["PROMETHEUS_METRICS_CACHE_UNAVAILABLE_THRESHOLD"]
    )

    validate_config(cfg, require_prometheus=require_prometheus)
    return cfg
