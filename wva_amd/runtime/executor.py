"""Polling executor with per-tick infinite retry.

Parity: reference internal/engines/executor/polling.go:28-83 — fixed
interval loop; each tick retries forever with exponential backoff capped
at 4s until the tick function succeeds or the executor is stopped.
"""
from __future__ import annotations

import threading
from typing import Callable, Optional

from ..utils.logging import get_logger

log = get_logger("runtime.executor")

MAX_RETRY_BACKOFF_SECONDS = 4.0
INITIAL_RETRY_BACKOFF_SECONDS = 0.25


class PollingExecutor:
    def __init__(
        self,
        interval_seconds: float,
        tick: Callable[[], None],
        name: str = "executor",
    ):
        self.interval_seconds = interval_seconds
        self.tick = tick
        self.name = name
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self) -> None:
        if self._thread is not None:
            return
        self._thread = threading.Thread(
            target=self._run, name=self.name, daemon=True
        )
        self._thread.start()

    def stop(self, timeout: float = 5.0) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=timeout)
            self._thread = None

    def run_once(self) -> None:
        """Execute exactly one tick (with retry) — used by tests and bench."""
        self._execute_with_retry()

    def _run(self) -> None:
        # First tick immediately (wait.UntilWithContext semantics), then
        # every interval.
        while not self._stop.is_set():
            self._execute_with_retry()
            if self._stop.wait(self.interval_seconds):
                return

    def _execute_with_retry(self) -> None:
        backoff = INITIAL_RETRY_BACKOFF_SECONDS
        while not self._stop.is_set():
            try:
                self.tick()
                return
            except Exception as e:  # noqa: BLE001 — infinite retry per tick
                log.error("%s tick failed (retrying in %.2fs): %s", self.name, backoff, e)
                if self._stop.wait(backoff):
                    return
                backoff = min(backoff * 2, MAX_RETRY_BACKOFF_SECONDS)
