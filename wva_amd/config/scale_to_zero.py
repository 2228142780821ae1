"""Per-model scale-to-zero configuration.

Parity: reference internal/config/scale_to_zero.go:17-225 — same YAML
schema (`enable_scale_to_zero`, `retention_period`, `model_id`,
`namespace`), "default" key inheritance, priority chain
(per-model > global default > WVA_SCALE_TO_ZERO env > false), 10-minute
default retention, and first-key-wins duplicate handling over sorted keys.
"""
from __future__ import annotations

import os
import re
from dataclasses import dataclass
from typing import Dict, Optional

import yaml

from ..utils.logging import get_logger

log = get_logger("config.scale_to_zero")

DEFAULT_SCALE_TO_ZERO_RETENTION_SECONDS = 10 * 60
GLOBAL_DEFAULTS_KEY = "default"

_DURATION_RE = re.compile(r"(\d+(?:\.\d+)?)(ns|us|µs|ms|s|m|h)")
_UNIT_SECONDS = {
    "ns": 1e-9,
    "us": 1e-6,
    "µs": 1e-6,
    "ms": 1e-3,
    "s": 1.0,
    "m": 60.0,
    "h": 3600.0,
}


def parse_go_duration(s: str) -> float:
    """Parse a Go-style duration string ("5m", "1h30m", "90s") to seconds."""
    if not s:
        raise ValueError("duration cannot be empty")
    s = s.strip()
    pos = 0
    total = 0.0
    sign = 1.0
    if s.startswith("-"):
        sign = -1.0
        pos = 1
    matched_any = False
    for m in _DURATION_RE.finditer(s, pos):
        if m.start() != pos:
            raise ValueError(f"invalid duration format: {s!r}")
        total += float(m.group(1)) * _UNIT_SECONDS[m.group(2)]
        pos = m.end()
        matched_any = True
    if not matched_any or pos != len(s):
        raise ValueError(f"invalid duration format: {s!r}")
    return sign * total


def validate_retention_period(retention_period: str) -> float:
    """Validate and parse a retention period; returns seconds or raises ValueError."""
    if not retention_period:
        raise ValueError("retention period cannot be empty")
    seconds = parse_go_duration(retention_period)
    if seconds <= 0:
        raise ValueError(f"retention period must be positive, got {retention_period}")
    if seconds > 24 * 3600:
        log.info(
            "retention period is unusually long: %s — consider a shorter period",
            retention_period,
        )
    return seconds


@dataclass
class ModelScaleToZeroConfig:
    model_id: str = ""
    namespace: str = ""
    # None = not set (inherit), True/False = explicit
    enable_scale_to_zero: Optional[bool] = None
    retention_period: str = ""


# model ID (or "default") → config
ScaleToZeroConfigData = Dict[str, ModelScaleToZeroConfig]


def is_scale_to_zero_enabled(
    config_data: ScaleToZeroConfigData, model_id: str
) -> bool:
    cfg = config_data.get(model_id)
    if cfg is not None and cfg.enable_scale_to_zero is not None:
        return cfg.enable_scale_to_zero
    global_cfg = config_data.get(GLOBAL_DEFAULTS_KEY)
    if global_cfg is not None and global_cfg.enable_scale_to_zero is not None:
        return global_cfg.enable_scale_to_zero
    return os.environ.get("WVA_SCALE_TO_ZERO", "").lower() == "true"


def scale_to_zero_retention_seconds(
    config_data: ScaleToZeroConfigData, model_id: str
) -> float:
    cfg = config_data.get(model_id)
    if cfg is not None and cfg.retention_period:
        try:
            return validate_retention_period(cfg.retention_period)
        except ValueError as e:
            log.info(
                "invalid retention period for model %s (%s), checking global defaults",
                model_id,
                e,
            )
    global_cfg = config_data.get(GLOBAL_DEFAULTS_KEY)
    if global_cfg is not None and global_cfg.retention_period:
        try:
            return validate_retention_period(global_cfg.retention_period)
        except ValueError as e:
            log.info("invalid global default retention period (%s), using default", e)
            return DEFAULT_SCALE_TO_ZERO_RETENTION_SECONDS
    return DEFAULT_SCALE_TO_ZERO_RETENTION_SECONDS


def min_num_replicas(config_data: ScaleToZeroConfigData, model_id: str) -> int:
    return 0 if is_scale_to_zero_enabled(config_data, model_id) else 1


def parse_scale_to_zero_configmap(
    data: Optional[Dict[str, str]],
) -> ScaleToZeroConfigData:
    """Parse the `wva-model-scale-to-zero-config` ConfigMap data section."""
    out: ScaleToZeroConfigData = {}
    if not data:
        return out
    model_id_to_keys: Dict[str, list] = {}
    for key in sorted(data.keys()):
        raw = data[key]
        try:
            parsed = yaml.safe_load(raw) or {}
            if not isinstance(parsed, dict):
                raise ValueError("entry must be a mapping")
            cfg = ModelScaleToZeroConfig(
                model_id=parsed.get("model_id", "") or "",
                namespace=parsed.get("namespace", "") or "",
                enable_scale_to_zero=(
                    None
                    if "enable_scale_to_zero" not in parsed
                    else bool(parsed["enable_scale_to_zero"])
                ),
                retention_period=parsed.get("retention_period", "") or "",
            )
        except Exception as e:  # noqa: BLE001 — parse failure skips the entry
            log.info("failed to parse scale-to-zero config entry %s: %s", key, e)
            continue
        if key == GLOBAL_DEFAULTS_KEY:
            out[GLOBAL_DEFAULTS_KEY] = cfg
            continue
        if not cfg.model_id:
            log.info("skipping scale-to-zero config without model_id field: %s", key)
            continue
        if cfg.model_id in model_id_to_keys:
            log.info(
                "duplicate model_id %s in scale-to-zero ConfigMap — first key wins",
                cfg.model_id,
            )
            continue
        model_id_to_keys[cfg.model_id] = [key]
        out[cfg.model_id] = cfg
    return out
