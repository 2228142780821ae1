"""Exponential-backoff retry helpers.

Parity: reference internal/utils/utils.go:34-121 (backoff presets and
generic GetResourceWithBackoff / QueryPrometheusWithBackoff wrappers).
"""
from __future__ import annotations

import time
from typing import Callable, Optional, Type, TypeVar

T = TypeVar("T")

# Preset mirroring the reference's k8s-read backoff (utils.go:34-47)
DEFAULT_MAX_ATTEMPTS = 4
DEFAULT_INITIAL_DELAY = 0.2
DEFAULT_FACTOR = 2.0
DEFAULT_MAX_DELAY = 4.0


def retry_with_backoff(
    fn: Callable[[], T],
    max_attempts: int = DEFAULT_MAX_ATTEMPTS,
    initial_delay: float = DEFAULT_INITIAL_DELAY,
    factor: float = DEFAULT_FACTOR,
    max_delay: float = DEFAULT_MAX_DELAY,
    retry_on: Type[BaseException] = Exception,
    sleep: Callable[[float], None] = time.sleep,
    should_retry: Optional[Callable[[BaseException], bool]] = None,
) -> T:
    """Run fn() retrying on exception with exponential backoff.

    Raises the last exception when attempts are exhausted.
    """
    delay = initial_delay
    last_exc: Optional[BaseException] = None
    for attempt in range(max_attempts):
        try:
            return fn()
        except retry_on as e:  # noqa: PERF203
            if should_retry is not None and not should_retry(e):
                raise
            last_exc = e
            if attempt + 1 < max_attempts:
                sleep(delay)
                delay = min(delay * factor, max_delay)
    assert last_exc is not None
    raise last_exc
