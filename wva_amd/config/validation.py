"""Startup config validation (reference internal/config/validation.go)."""
from __future__ import annotations

import re
from urllib.parse import urlparse


class ConfigLoadError(ValueError):
    pass


_BIND_ADDR_RE = re.compile(r"^(0|[\w.\-]*:\d+)$")


def validate_config(cfg, require_prometheus: bool = True) -> None:
    from .config import Config  # local import to avoid cycle

    assert isinstance(cfg, Config)
    errs = []
    infra = cfg.infra
    if not _BIND_ADDR_RE.match(infra.metrics_bind_address or "0"):
        errs.append(
            f"METRICS_BIND_ADDRESS invalid: {infra.metrics_bind_address!r}"
        )
    if not _BIND_ADDR_RE.match(infra.health_probe_bind_address or "0"):
        errs.append(
            f"HEALTH_PROBE_BIND_ADDRESS invalid: {infra.health_probe_bind_address!r}"
        )
    if infra.lease_duration_seconds <= infra.renew_deadline_seconds:
        errs.append(
            "LEADER_ELECTION_LEASE_DURATION must be > LEADER_ELECTION_RENEW_DEADLINE"
        )
    if infra.renew_deadline_seconds <= infra.retry_period_seconds:
        errs.append(
            "LEADER_ELECTION_RENEW_DEADLINE must be > LEADER_ELECTION_RETRY_PERIOD"
        )
    if infra.optimization_interval_seconds <= 0:
        errs.append("GLOBAL_OPT_INTERVAL must be positive")
    if infra.rest_client_timeout_seconds <= 0:
        errs.append("REST_CLIENT_TIMEOUT must be positive")

    if require_prometheus:
        if not cfg.prometheus.base_url:
            errs.append("PROMETHEUS_BASE_URL is required")
        else:
            parsed = urlparse(cfg.prometheus.base_url)
            if parsed.scheme not in ("http", "https") or not parsed.netloc:
                errs.append(
                    f"PROMETHEUS_BASE_URL must be an http(s) URL, got "
                    f"{cfg.prometheus.base_url!r}"
                )
    if cfg.cache.ttl_seconds <= 0:
        errs.append("PROMETHEUS_METRICS_CACHE_TTL must be positive")
    if errs:
        raise ConfigLoadError("; ".join(errs))
