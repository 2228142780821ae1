"""Datastore: InferencePool registry + namespace tracking.

Parity: reference internal/datastore/datastore.go:39-260 — holds the
EndpointPools converted from InferencePool resources, instantiates a
PodScrapingSource per pool, and tracks namespaces that contain VAs (for
namespace-local ConfigMap watching).
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional

from ..collector.pod_scraping_source import FetchFunc, PodScrapingSource
from ..kube.fake import FakeCluster
from ..kube.objects import EndpointPool


class Datastore:
    def __init__(
        self,
        cluster: FakeCluster,
        epp_bearer_token: str = "",
        epp_metrics_reader_secret_name: str = "",
        epp_metrics_reader_secret_key: str = "token",
        scrape_fetch: Optional[FetchFunc] = None,
        source_registry=None,
    ):
        self.cluster = cluster
        self.epp_bearer_token = epp_bearer_token
        self.epp_metrics_reader_secret_name = epp_metrics_reader_secret_name
        self.epp_metrics_reader_secret_key = epp_metrics_reader_secret_key
        self.scrape_fetch = scrape_fetch
        # Optional collector SourceRegistry: pool sources register there
        # under their pool name (reference datastore.go registers one
        # PodScrapingSource per InferencePool in the registry)
        self.source_registry = source_registry
        self._lock = threading.RLock()
        self._pools: Dict[str, EndpointPool] = {}
        self._pool_sources: Dict[str, PodScrapingSource] = {}
        self._tracked_namespaces: Dict[str, bool] = {}
        self._watched_namespaces: Dict[str, bool] = {}

    @staticmethod
    def _pool_key(namespace: str, name: str) -> str:
        return f"{namespace}/{name}"

    # --- pools ---

    def pool_set(self, pool: EndpointPool) -> None:
        with self._lock:
            key = self._pool_key(pool.namespace, pool.name)
            self._pools[key] = pool
            source = PodScrapingSource(
                self.cluster,
                pool,
                bearer_token=self.epp_bearer_token,
                metrics_reader_secret_name=self.epp_metrics_reader_secret_name,
                metrics_reader_secret_key=self.epp_metrics_reader_secret_key,
                fetch=self.scrape_fetch,
            )
            self._pool_sources[key] = source
            if self.source_registry is not None:
                self.source_registry.register(source)

    def pool_get(self, namespace: str, name: str) -> Optional[EndpointPool]:
        with self._lock:
            return self._pools.get(self._pool_key(namespace, name))

    def pool_delete(self, namespace: str, name: str) -> None:
        with self._lock:
            key = self._pool_key(namespace, name)
            self._pools.pop(key, None)
            source = self._pool_sources.pop(key, None)
            if source is not None and self.source_registry is not None:
                self.source_registry.unregister(source.name())

    def pool_source(
        self, namespace: str, name: str
    ) -> Optional[PodScrapingSource]:
        with self._lock:
            return self._pool_sources.get(self._pool_key(namespace, name))

    def pool_get_from_labels(
        self, namespace: str, labels: Dict[str, str]
    ) -> Optional[EndpointPool]:
        """Find the pool whose selector matches the given pod-template
        labels (scale-from-zero target resolution)."""
        with self._lock:
            for pool in self._pools.values():
                if pool.namespace != namespace:
                    continue
                if pool.selector and all(
                    labels.get(k) == v for k, v in pool.selector.items()
                ):
                    return pool
            return None

    def pools(self) -> List[EndpointPool]:
        with self._lock:
            return list(self._pools.values())

    # --- namespace tracking ---

    def namespace_track(self, namespace: str) -> None:
        with self._lock:
            self._tracked_namespaces[namespace] = True

    def namespace_untrack(self, namespace: str) -> None:
        with self._lock:
            self._tracked_namespaces.pop(namespace, None)

    def namespace_is_tracked(self, namespace: str) -> bool:
        with self._lock:
            return namespace in self._tracked_namespaces

    def namespace_watch(self, namespace: str) -> None:
        with self._lock:
            self._watched_namespaces[namespace] = True

    def namespace_is_watched(self, namespace: str) -> bool:
        with self._lock:
            return namespace in self._watched_namespaces

    def tracked_namespaces(self) -> List[str]:
        with self._lock:
            return sorted(self._tracked_namespaces)
