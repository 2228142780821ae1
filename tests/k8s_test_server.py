"""Minimal in-process Kubernetes API server for exercising RestCluster.

Serves the REST subset kube/rest.py uses — list/get/create/put/
merge-patch (status, scale)/delete/watch (chunked JSON stream) — backed
by a FakeCluster through the same serde layer, so the REST client is
tested against API-server-shaped HTTP without a network. The envtest
analog for the REST path.
"""
from __future__ import annotations

import json
import re
import threading
import queue
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional, Tuple
from urllib.parse import parse_qs, urlparse

from wva_amd.kube import serde
from wva_amd.kube.fake import ConflictError, FakeCluster, NotFoundError

# path prefix → kind (built from the serde table)
_ROUTES = []
for kind, (_, _, (prefix, plural, namespaced)) in serde.SERDE.items():
    _ROUTES.append((prefix, plural, namespaced, kind))


def _match(path: str) -> Optional[Tuple[str, Optional[str], Optional[str], str]]:
    """→ (kind, namespace, name, subresource) or None."""
    for prefix, plural, namespaced, kind in _ROUTES:
        if namespaced:
            m = re.fullmatch(
                f"/{prefix}/namespaces/(?P<ns>[^/]+)/{plural}"
                f"(?:/(?P<name>[^/]+))?(?:/(?P<sub>status|scale))?",
                path,
            )
            if m:
                return kind, m.group("ns"), m.group("name"), m.group("sub") or ""
        m = re.fullmatch(
            f"/{prefix}/{plural}(?:/(?P<name>[^/]+))?(?:/(?P<sub>status|scale))?",
            path,
        )
        if m:
            return kind, None, m.group("name"), m.group("sub") or ""
    return None


class K8sTestServer:
    def __init__(self, cluster: Optional[FakeCluster] = None):
        self.cluster = cluster if cluster is not None else FakeCluster()
        outer = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, *a):
                pass

            def _send(self, code: int, payload: dict):
                body = json.dumps(payload).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def _body(self) -> dict:
                n = int(self.headers.get("Content-Length") or 0)
                return json.loads(self.rfile.read(n)) if n else {}

            def do_GET(self):
                parsed = urlparse(self.path)
                qs = parse_qs(parsed.query)
                route = _match(parsed.path)
                if route is None:
                    return self._send(404, {"message": "no route"})
                kind, ns, name, _sub = route
                if name:
                    obj = outer.cluster.try_get(kind, ns or "", name)
                    if obj is None:
                        return self._send(404, {"message": "not found"})
                    return self._send(200, serde.encode(obj))
                if qs.get("watch", ["false"])[0] == "true":
                    rv = int(qs.get("resourceVersion", ["-1"])[0] or -1)
                    return self._watch(kind, rv)
                sel = None
                if "labelSelector" in qs:
                    sel = dict(
                        kv.split("=", 1) for kv in qs["labelSelector"][0].split(",")
                    )
                objs, snap_rv = outer.cluster.snapshot(
                    kind, namespace=ns, label_selector=sel
                )
                return self._send(200, {
                    "apiVersion": "v1",
                    "kind": f"{kind}List",
                    "metadata": {"resourceVersion": str(snap_rv)},
                    "items": [serde.encode(o) for o in objs],
                })

            def _watch(self, kind: str, rv: int = -1):
                # rv >= 0: replay from the event log (atomic with the
                # subscription — the race a real API server closes with
                # its watch cache); rv < 0: live-only
                if rv >= 0:
                    q = outer.cluster.watch_since([kind], rv)
                else:
                    q = outer.cluster.watch([kind])
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Transfer-Encoding", "chunked")
                self.end_headers()
                try:
                    while True:
                        try:
                            evt = q.get(timeout=0.2)
                        except queue.Empty:
                            if outer._closing.is_set():
                                break
                            continue
                        line = json.dumps({
                            "type": evt.type,
                            "object": serde.encode(evt.obj),
                        }).encode() + b"\n"
                        self.wfile.write(f"{len(line):x}\r\n".encode())
                        self.wfile.write(line + b"\r\n")
                        self.wfile.flush()
                except (BrokenPipeError, ConnectionResetError):
                    pass
                finally:
                    outer.cluster.stop_watch(q)

            def do_POST(self):
                route = _match(urlparse(self.path).path)
                if route is None:
                    # events endpoint: accept silently
                    if "/events" in self.path:
                        return self._send(201, {})
                    return self._send(404, {"message": "no route"})
                kind, ns, _name, _sub = route
                body = self._body()
                obj = serde.decode(kind, body)
                try:
                    created = outer.cluster.create(obj)
                except ConflictError as e:
                    return self._send(409, {"message": str(e)})
                return self._send(201, serde.encode(created))

            def do_PUT(self):
                route = _match(urlparse(self.path).path)
                if route is None:
                    return self._send(404, {"message": "no route"})
                kind, ns, name, _sub = route
                obj = serde.decode(kind, self._body())
                try:
                    updated = outer.cluster.update(obj)
                except NotFoundError:
                    return self._send(404, {"message": "not found"})
                return self._send(200, serde.encode(updated))

            def do_PATCH(self):
                route = _match(urlparse(self.path).path)
                if route is None:
                    return self._send(404, {"message": "no route"})
                kind, ns, name, sub = route
                body = self._body()
                if sub == "scale":
                    replicas = int(
                        ((body.get("spec") or {}).get("replicas", 0))
                    )
                    try:
                        obj = outer.cluster.scale(kind, ns or "", name, replicas)
                    except NotFoundError:
                        return self._send(404, {"message": "not found"})
                    return self._send(200, serde.encode(obj))
                if sub == "status":
                    obj = serde.decode(kind, body)
                    try:
                        updated = outer.cluster.update_status(obj)
                    except NotFoundError:
                        return self._send(404, {"message": "not found"})
                    return self._send(200, serde.encode(updated))
                return self._send(400, {"message": "unsupported patch"})

            def do_DELETE(self):
                route = _match(urlparse(self.path).path)
                if route is None:
                    return self._send(404, {"message": "no route"})
                kind, ns, name, _sub = route
                try:
                    outer.cluster.delete(kind, ns or "", name)
                except NotFoundError:
                    return self._send(404, {"message": "not found"})
                return self._send(200, {"status": "Success"})

        self._closing = threading.Event()
        self.server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self._thread: Optional[threading.Thread] = None

    @property
    def url(self) -> str:
        host, port = self.server.server_address
        return f"http://{host}:{port}"

    def start(self) -> "K8sTestServer":
        self._thread = threading.Thread(
            target=self.server.serve_forever, daemon=True, name="k8s-test-api"
        )
        self._thread.start()
        return self

    def stop(self) -> None:
        self._closing.set()
        self.server.shutdown()
        self.server.server_close()
        if self._thread:
            self._thread.join(timeout=5)
