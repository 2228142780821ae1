"""Manager — the controller-runtime analog.

Parity: reference cmd/main.go:83-520 wiring — a Manager owns watch-driven
reconcilers, leader-election-gated runnables (the engines), health/ready
probes, and the DecisionTrigger consumer feeding the VA reconciler.

Structure: one dispatcher thread drains cluster watch events through
per-registration predicates into a work queue; one worker thread executes
reconciles (dedup in the queue); a trigger thread drains the
DecisionTrigger. Engines are Runnables started only while this manager
holds the leader Lease (or immediately when leader election is off).
"""
from __future__ import annotations

import queue
import threading
import time
import uuid
from dataclasses import dataclass
from typing import Callable, List, Optional, Protocol, Tuple

from ..config.config import Config
from ..engines.common import DecisionTrigger
from ..kube.fake import FakeCluster, NotFoundError, WatchEvent
from ..kube.objects import Lease
from ..api.types import ObjectMeta
from ..utils.logging import get_logger

log = get_logger("runtime.manager")

ReconcileFunc = Callable[[str, str], None]  # (namespace, name)
MapFunc = Callable[[WatchEvent], Optional[Tuple[str, str]]]


class DedupWorkQueue:
    """controller-runtime workqueue semantics: an item already queued
    is not queued again (dedup while PENDING); once a worker picks it
    up it may be re-added — a burst of watch events for one object
    collapses into at most one queued + one in-flight reconcile."""

    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._cond = threading.Condition(self._lock)
        self._order: List[Tuple[ReconcileFunc, str, str]] = []
        self._pending: set = set()

    def put(self, item: Tuple[ReconcileFunc, str, str]) -> None:
        with self._cond:
            if item in self._pending:
                return
            self._pending.add(item)
            self._order.append(item)
            self._cond.notify()

    def get(self, timeout: float) -> Optional[Tuple[ReconcileFunc, str, str]]:
        with self._cond:
            if not self._order:
                self._cond.wait(timeout)
            if not self._order:
                return None
            item = self._order.pop(0)
            self._pending.discard(item)
            return item

    def empty(self) -> bool:
        with self._lock:
            return not self._order


class Runnable(Protocol):
    def start(self) -> None: ...
    def stop(self) -> None: ...


@dataclass
class _Registration:
    kinds: List[str]
    predicate: Callable[[WatchEvent], bool]
    map_func: MapFunc
    reconcile: ReconcileFunc


def identity_map(event: WatchEvent) -> Optional[Tuple[str, str]]:
    return (event.obj.metadata.namespace, event.obj.metadata.name)


class LeaderElector:
    """Lease-based leader election (cmd/main.go:266-287 semantics:
    tuned timings + release-on-cancel fast failover)."""

    LEASE_NAMESPACE = "kube-system"

    def __init__(
        self,
        cluster: FakeCluster,
        lease_name: str,
        identity: Optional[str] = None,
        lease_duration: float = 60.0,
        renew_deadline: float = 50.0,
        retry_period: float = 10.0,
    ):
        self.cluster = cluster
        self.lease_name = lease_name
        self.identity = identity or f"wva-{uuid.uuid4().hex[:8]}"
        self.lease_duration = lease_duration
        self.renew_deadline = renew_deadline
        self.retry_period = retry_period

    def try_acquire_or_renew(self) -> bool:
        now = time.time()
        lease = self.cluster.try_get("Lease", self.LEASE_NAMESPACE, self.lease_name)
        if lease is None:
            lease = Lease(
                metadata=ObjectMeta(
                    name=self.lease_name, namespace=self.LEASE_NAMESPACE
                ),
                holder_identity=self.identity,
                lease_duration_seconds=self.lease_duration,
                acquire_time=now,
                renew_time=now,
            )
            try:
                self.cluster.create(lease)
                return True
            except Exception:  # noqa: BLE001 — lost the race
                return False
        from ..kube.fake import ConflictError

        if lease.holder_identity == self.identity:
            lease.renew_time = now
            try:
                self.cluster.update(lease)
            except ConflictError:
                # a competitor wrote between our read and renew — the
                # optimistic-concurrency loss IS the election loss
                return False
            return True
        expired = (
            lease.renew_time is None
            or now - lease.renew_time > lease.lease_duration_seconds
        )
        if expired:
            lease.holder_identity = self.identity
            lease.acquire_time = now
            lease.renew_time = now
            try:
                self.cluster.update(lease)
            except ConflictError:
                return False
            return True
        return False

    def release(self) -> None:
        # best-effort (ReleaseOnCancel): failure here must never abort
        # shutdown — a competitor writing the lease concurrently
        # (ConflictError) or an unreachable API server just means the
        # lease expires on its own
        try:
            lease = self.cluster.try_get(
                "Lease", self.LEASE_NAMESPACE, self.lease_name
            )
            if lease is not None and lease.holder_identity == self.identity:
                lease.holder_identity = ""
                lease.renew_time = None
                self.cluster.update(lease)
        except Exception as e:  # noqa: BLE001
            log.debug("lease release failed (ignored): %s", e)


class Manager:
    def __init__(
        self,
        cluster: FakeCluster,
        config: Config,
        decision_trigger: Optional[DecisionTrigger] = None,
    ):
        self.cluster = cluster
        self.config = config
        self.decision_trigger = decision_trigger
        self._registrations: List[_Registration] = []
        self._runnables: List[Runnable] = []
        self._va_reconcile: Optional[ReconcileFunc] = None
        self._work = DedupWorkQueue()
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._watch_q: Optional["queue.Queue[WatchEvent]"] = None
        self._started = False
        self.elector: Optional[LeaderElector] = None
        self._is_leader = threading.Event()
        self._runnables_started = False
        self._health_checks: List[Callable[[], bool]] = []
        self._ready_checks: List[Callable[[], bool]] = [
            lambda: self.config.is_bootstrap_complete()
        ]

    # --- registration ---

    def register_reconciler(
        self,
        kinds: List[str],
        predicate: Callable[[WatchEvent], bool],
        reconcile: ReconcileFunc,
        map_func: MapFunc = identity_map,
        is_va_reconciler: bool = False,
    ) -> None:
        self._registrations.append(
            _Registration(kinds, predicate, map_func, reconcile)
        )
        if is_va_reconciler:
            self._va_reconcile = reconcile

    def add_runnable(self, runnable: Runnable) -> None:
        self._runnables.append(runnable)

    def add_healthz_check(self, check: Callable[[], bool]) -> None:
        self._health_checks.append(check)

    def add_readyz_check(self, check: Callable[[], bool]) -> None:
        self._ready_checks.append(check)

    def healthz(self) -> bool:
        return all(c() for c in self._health_checks)

    def readyz(self) -> bool:
        """Gated on ConfigMap bootstrap (cmd/main.go:486-498)."""
        return all(c() for c in self._ready_checks)

    def is_leader(self) -> bool:
        return self._is_leader.is_set()

    # --- lifecycle ---

    def start(self) -> None:
        if self._started:
            return
        self._started = True
        kinds = sorted({k for r in self._registrations for k in r.kinds})
        self._watch_q = self.cluster.watch(kinds)

        self._spawn(self._dispatch_loop, "mgr-dispatch")
        self._spawn(self._worker_loop, "mgr-worker")
        if self.decision_trigger is not None:
            self._spawn(self._trigger_loop, "mgr-trigger")

        infra = self.config.infra
        if infra.enable_leader_election:
            self.elector = LeaderElector(
                self.cluster,
                infra.leader_election_id,
                lease_duration=infra.lease_duration_seconds,
                renew_deadline=infra.renew_deadline_seconds,
                retry_period=infra.retry_period_seconds,
            )
            self._spawn(self._election_loop, "mgr-election")
        else:
            self._is_leader.set()
            self._start_runnables()

    def stop(self) -> None:
        self._stop.set()
        for r in self._runnables:
            try:
                r.stop()
            except Exception:  # noqa: BLE001
                pass
        if self.elector is not None:
            self.elector.release()  # ReleaseOnCancel fast failover
        if self._watch_q is not None:
            self.cluster.stop_watch(self._watch_q)
            self._watch_q.put(None)  # type: ignore[arg-type]  # unblock
        for t in self._threads:
            t.join(timeout=2.0)
        self._threads.clear()

    def _spawn(self, target, name: str) -> None:
        t = threading.Thread(target=target, name=name, daemon=True)
        t.start()
        self._threads.append(t)

    def _start_runnables(self) -> None:
        if self._runnables_started:
            return
        self._runnables_started = True
        for r in self._runnables:
            r.start()

    # --- loops ---

    def _election_loop(self) -> None:
        assert self.elector is not None
        while not self._stop.is_set():
            acquired = self.elector.try_acquire_or_renew()
            if acquired and not self._is_leader.is_set():
                log.info("acquired leadership (%s)", self.elector.identity)
                self._is_leader.set()
                self._start_runnables()
            elif not acquired and self._is_leader.is_set():
                log.info("lost leadership (%s)", self.elector.identity)
                self._is_leader.clear()
                for r in self._runnables:
                    r.stop()
                self._runnables_started = False
            self._stop.wait(self.elector.retry_period)

    def _dispatch_loop(self) -> None:
        assert self._watch_q is not None
        while not self._stop.is_set():
            try:
                event = self._watch_q.get(timeout=0.2)
            except queue.Empty:
                continue
            if event is None:
                return
            if event.type not in ("ADDED", "MODIFIED", "DELETED"):
                continue  # SYNC/BOOKMARK markers from cache/REST layers
            for r in self._registrations:
                if event.kind not in r.kinds:
                    continue
                try:
                    if not r.predicate(event):
                        continue
                    target = r.map_func(event)
                    if target is not None:
                        self._work.put((r.reconcile, target[0], target[1]))
                except Exception as e:  # noqa: BLE001
                    log.error("dispatch error for %s: %s", event.kind, e)

    def _worker_loop(self) -> None:
        while not self._stop.is_set():
            item = self._work.get(timeout=0.2)
            if item is None:
                continue
            reconcile, ns, name = item
            try:
                reconcile(ns, name)
            except Exception as e:  # noqa: BLE001
                log.error("reconcile %s/%s failed: %s", ns, name, e)

    def _trigger_loop(self) -> None:
        assert self.decision_trigger is not None
        while not self._stop.is_set():
            key = self.decision_trigger.pop(timeout=0.2)
            if key is None:
                continue
            if self._va_reconcile is None:
                continue
            ns, _, name = key.partition("/")
            self._work.put((self._va_reconcile, ns, name))

    # --- test/bench helper ---

    def drain(self, timeout: float = 2.0) -> None:
        """Wait until the work queue is empty (best effort)."""
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if self._work.empty() and (
                self.decision_trigger is None or len(self.decision_trigger) == 0
            ):
                time.sleep(0.05)
                if self._work.empty():
                    return
            time.sleep(0.02)
