"""Probe + metrics HTTP endpoints.

The reference binds two listeners from cmd/main.go: a metrics address
(controller-runtime /metrics, reference cmd/main.go:266-287) and a health
probe address serving /healthz + /readyz, with readyz gated on ConfigMap
bootstrap (cmd/main.go:482-498). This module is the Python equivalent:
one stdlib ThreadingHTTPServer per address, run as daemon threads.

/metrics renders the prometheus_client registry the MetricsEmitter
registered its wva_* series into — the HPA/prometheus-adapter contract
surface.
"""
from __future__ import annotations

import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Callable, Optional, Tuple

from prometheus_client import CollectorRegistry, REGISTRY, generate_latest
from prometheus_client.exposition import CONTENT_TYPE_LATEST


def _parse_bind(addr: str) -> Tuple[str, int]:
    """':8081' / '0.0.0.0:8081' / '8081' → (host, port)."""
    addr = addr.strip()
    if ":" in addr:
        host, _, port = addr.rpartition(":")
        return host or "0.0.0.0", int(port)
    return "0.0.0.0", int(addr)


class ProbeServer:
    """Serves /healthz, /readyz and (optionally) /metrics on one address."""

    def __init__(
        self,
        bind_address: str,
        healthz: Callable[[], bool],
        readyz: Callable[[], bool],
        registry: Optional[CollectorRegistry] = None,
        serve_metrics: bool = True,
        cert_watcher: Optional["CertWatcher"] = None,
    ):
        self._healthz = healthz
        self._readyz = readyz
        self._registry = registry if registry is not None else REGISTRY
        self._serve_metrics = serve_metrics
        host, port = _parse_bind(bind_address)
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def do_GET(self):  # noqa: N802 — stdlib handler API
                if self.path.startswith("/healthz"):
                    self._probe(outer._healthz)
                elif self.path.startswith("/readyz"):
                    self._probe(outer._readyz)
                elif self.path.startswith("/metrics") and outer._serve_metrics:
                    body = generate_latest(outer._registry)
                    self.send_response(200)
                    self.send_header("Content-Type", CONTENT_TYPE_LATEST)
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)
                elif self.path.startswith("/debug/threads"):
                    # pprof-analog (SURVEY §5 "optionally add pprof"):
                    # live stack dump of every thread — the first tool
                    # for a wedged dispatcher/engine in production
                    import sys as _sys
                    import threading as _threading
                    import traceback as _traceback

                    lines = []
                    names = {
                        t.ident: t.name for t in _threading.enumerate()
                    }
                    for tid, frame in _sys._current_frames().items():
                        lines.append(
                            f"--- thread {names.get(tid, '?')} ({tid}) ---"
                        )
                        lines.extend(
                            l.rstrip() for l in
                            _traceback.format_stack(frame)
                        )
                    body = ("\n".join(lines) + "\n").encode()
                    self.send_response(200)
                    self.send_header("Content-Type", "text/plain")
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)
                else:
                    self.send_response(404)
                    self.end_headers()

            def _probe(self, check: Callable[[], bool]) -> None:
                try:
                    ok = check()
                except Exception:  # noqa: BLE001 — probe must answer
                    ok = False
                body = b"ok" if ok else b"unhealthy"
                self.send_response(200 if ok else 503)
                self.send_header("Content-Type", "text/plain")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *args):  # silence per-request lines
                pass

        self._server = ThreadingHTTPServer((host, port), Handler)
        self._cert_watcher = cert_watcher
        if cert_watcher is not None:
            # TLS listener with hot-reloaded certs (metrics-over-TLS,
            # cmd/main.go:156-170 disables HTTP/2 — stdlib http.server
            # is HTTP/1.1-only, so that mitigation is structural here)
            self._server.socket = cert_watcher.ssl_context.wrap_socket(
                self._server.socket, server_side=True
            )
        self._thread: Optional[threading.Thread] = None

    @property
    def port(self) -> int:
        return self._server.server_address[1]

    def start(self) -> None:
        if self._cert_watcher is not None:
            self._cert_watcher.start()
        self._thread = threading.Thread(
            target=self._server.serve_forever, daemon=True, name="probe-http"
        )
        self._thread.start()

    def stop(self) -> None:
        if self._cert_watcher is not None:
            self._cert_watcher.stop()
        self._server.shutdown()
        self._server.server_close()
        if self._thread:
            self._thread.join(timeout=5)


class CertWatcher:
    """Certificate hot-reload for a TLS listener (reference
    cmd/main.go:172-249: metrics/webhook cert watchers).

    Polls the cert/key files' mtimes and reloads the chain INTO THE
    SAME SSLContext when they change — the listening socket binds the
    context object at wrap time, so new handshakes pick the rotated
    certs up without rebinding; a context swap would silently keep
    serving the old ones. Failed reloads keep the previous chain.
    """

    def __init__(self, cert_path: str, key_path: str,
                 poll_seconds: float = 30.0):
        import ssl

        self.cert_path = cert_path
        self.key_path = key_path
        self.poll_seconds = poll_seconds
        self._mtimes = self._stat()
        self._ctx = self._load()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def _stat(self):
        import os

        try:
            return (os.stat(self.cert_path).st_mtime,
                    os.stat(self.key_path).st_mtime)
        except OSError:
            return (0.0, 0.0)

    def _load(self):
        import ssl

        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        ctx.load_cert_chain(self.cert_path, self.key_path)
        return ctx

    @property
    def ssl_context(self):
        return self._ctx

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self._watch, daemon=True, name="cert-watcher"
        )
        self._thread.start()

    def _watch(self) -> None:
        while not self._stop.wait(self.poll_seconds):
            mt = self._stat()
            if mt != self._mtimes:
                try:
                    # reload in place — the listener holds this context
                    self._ctx.load_cert_chain(self.cert_path, self.key_path)
                    self._mtimes = mt
                except Exception:  # noqa: BLE001 — keep old certs on error
                    pass

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)
