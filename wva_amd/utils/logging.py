"""Structured logging with the reference's verbosity ladder.

Parity: reference internal/logging/logger.go:13-38 — verbosity levels
DEFAULT=2, VERBOSE=3, DEBUG=4, TRACE=5 map onto Python logging levels.
"""
from __future__ import annotations

import logging
import os
import sys

DEFAULT = 2
VERBOSE = 3
DEBUG = 4
TRACE = 5

_LEVEL_MAP = {
    0: logging.WARNING,
    1: logging.INFO,
    2: logging.INFO,
    3: logging.DEBUG,
    4: logging.DEBUG,
    5: logging.DEBUG,
}

_configured = False


def setup_logging(verbosity: int = DEFAULT, stream=None) -> None:
    global _configured
    root = logging.getLogger("wva")
    root.setLevel(_LEVEL_MAP.get(verbosity, logging.INFO))
    if not _configured:
        handler = logging.StreamHandler(stream or sys.stderr)
        handler.setFormatter(
            logging.Formatter(
                "%(asctime)s %(levelname)-5s %(name)s: %(message)s",
                datefmt="%Y-%m-%dT%H:%M:%S",
            )
        )
        root.addHandler(handler)
        root.propagate = False
        _configured = True


def get_logger(name: str) -> logging.Logger:
    if not _configured:
        setup_logging(int(os.environ.get("WVA_LOG_VERBOSITY", DEFAULT)))
    return logging.getLogger(f"wva.{name}")
