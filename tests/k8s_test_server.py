"""In-process Kubernetes API server for exercising RestCluster.

The envtest analog for the REST path: serves the REST subset
kube/rest.py uses — list/get/create/put/merge-patch (status, scale)/
delete/watch (chunked JSON stream) — backed by a FakeCluster through
the same serde layer, with the API-server behaviors a homemade fake is
usually missing:

* **Optimistic concurrency**: PUT with a stale metadata.resourceVersion
  → 409 Conflict with a kube-shaped Status body.
* **CRD openAPI validation**: VariantAutoscaling create/update and the
  *result* of status merge patches are validated against the generated
  CRD schema (wva_amd/api/crd.py — the byte-parity schema), so a
  partial `desiredOptimizedAlloc` merge patch on a fresh VA is rejected
  with 422 exactly like reference issue #731
  (variantautoscaling_controller.go:237-252).
* **Merge-patch semantics**: PATCH .../status applies an RFC 7386 merge
  to the stored object's status (not a whole-status replace), then
  validates.
* **Watch bookmarks**: BOOKMARK events are sent periodically when the
  client asks allowWatchBookmarks=true.
* **Fault injection**: `drop_watches()` force-closes every active watch
  stream (client must reconnect from its last resourceVersion);
  `fail_next(n, code)` makes the next n mutating requests fail.
"""
from __future__ import annotations

import json
import re
import threading
import queue
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional, Tuple
from urllib.parse import parse_qs, urlparse

from wva_amd.api.crd import variantautoscaling_crd
from wva_amd.kube import serde
from wva_amd.kube.fake import (
    ConflictError, ExpiredError, FakeCluster, NotFoundError,
)
from wva_amd.kube.openapi import merge_patch, validate

# path prefix → kind (built from the serde table)
_ROUTES = []
for kind, (_, _, (prefix, plural, namespaced)) in serde.SERDE.items():
    _ROUTES.append((prefix, plural, namespaced, kind))

_VA_SCHEMA = variantautoscaling_crd()["spec"]["versions"][0]["schema"][
    "openAPIV3Schema"
]


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


def _status_body(code: int, reason: str, message: str) -> dict:
    """kube-apiserver metav1.Status shape."""
    return {
        "apiVersion": "v1",
        "kind": "Status",
        "status": "Failure",
        "message": message,
        "reason": reason,
        "code": code,
    }


class K8sTestServer:
    def __init__(self, cluster: Optional[FakeCluster] = None,
                 bookmark_interval_s: float = 0.5):
        self.cluster = cluster if cluster is not None else FakeCluster()
        self.bookmark_interval_s = bookmark_interval_s
        # fault injection state
        self._fail_lock = threading.Lock()
        self._fail_remaining = 0
        self._fail_code = 500
        # active watch bookkeeping for drop_watches()
        self._watch_lock = threading.Lock()
        self._watch_drops: list = []  # threading.Event per active stream
        self.request_counts: dict = {}  # method → count (observability)
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

            def _count(self, method: str) -> None:
                with outer._fail_lock:
                    outer.request_counts[method] = (
                        outer.request_counts.get(method, 0) + 1
                    )

            def _maybe_fail(self) -> bool:
                """Consume one injected failure; True if this request
                should fail."""
                with outer._fail_lock:
                    if outer._fail_remaining > 0:
                        outer._fail_remaining -= 1
                        code = outer._fail_code
                    else:
                        return False
                self._send(code, _status_body(
                    code, "InternalError", "injected failure"
                ))
                return True

            def do_GET(self):
                self._count("GET")
                parsed = urlparse(self.path)
                qs = parse_qs(parsed.query)
                route = _match(parsed.path)
                if route is None:
                    return self._send(404, _status_body(
                        404, "NotFound", "no route"
                    ))
                kind, ns, name, _sub = route
                if name:
                    obj = outer.cluster.try_get(kind, ns or "", name)
                    if obj is None:
                        return self._send(404, _status_body(
                            404, "NotFound", f"{kind} {name} not found"
                        ))
                    return self._send(200, serde.encode(obj))
                if qs.get("watch", ["false"])[0] == "true":
                    rv = int(qs.get("resourceVersion", ["-1"])[0] or -1)
                    bookmarks = (
                        qs.get("allowWatchBookmarks", ["false"])[0] == "true"
                    )
                    return self._watch(kind, rv, bookmarks)
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

            def _watch(self, kind: str, rv: int = -1, bookmarks: bool = False):
                # rv >= 0: replay from the event log (atomic with the
                # subscription — the race a real API server closes with
                # its watch cache); rv < 0: live-only
                if rv >= 0:
                    try:
                        q = outer.cluster.watch_since([kind], rv)
                    except ExpiredError as e:
                        return self._send(410, _status_body(
                            410, "Expired", str(e)
                        ))
                else:
                    q = outer.cluster.watch([kind])
                drop = threading.Event()
                with outer._watch_lock:
                    outer._watch_drops.append(drop)
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Transfer-Encoding", "chunked")
                self.end_headers()
                last_rv = max(rv, 0)
                last_bookmark = 0.0
                import time as _time

                def _chunk(payload: dict) -> None:
                    line = json.dumps(payload).encode() + b"\n"
                    self.wfile.write(f"{len(line):x}\r\n".encode())
                    self.wfile.write(line + b"\r\n")
                    self.wfile.flush()

                try:
                    while True:
                        try:
                            evt = q.get(timeout=0.1)
                        except queue.Empty:
                            if outer._closing.is_set() or drop.is_set():
                                break
                            now = _time.monotonic()
                            if (
                                bookmarks
                                and now - last_bookmark
                                > outer.bookmark_interval_s
                            ):
                                last_bookmark = now
                                _chunk({
                                    "type": "BOOKMARK",
                                    "object": {
                                        "kind": kind,
                                        "metadata": {
                                            "resourceVersion": str(
                                                max(last_rv, outer.cluster._rv)
                                            )
                                        },
                                    },
                                })
                            continue
                        if drop.is_set():
                            break
                        enc = serde.encode(evt.obj)
                        last_rv = int(
                            (enc.get("metadata") or {}).get(
                                "resourceVersion", last_rv
                            )
                        )
                        _chunk({"type": evt.type, "object": enc})
                except (BrokenPipeError, ConnectionResetError):
                    pass
                finally:
                    outer.cluster.stop_watch(q)
                    with outer._watch_lock:
                        if drop in outer._watch_drops:
                            outer._watch_drops.remove(drop)

            def do_POST(self):
                self._count("POST")
                if self._maybe_fail():
                    return
                route = _match(urlparse(self.path).path)
                if route is None:
                    # events endpoint: accept silently
                    if "/events" in self.path:
                        return self._send(201, {})
                    return self._send(404, _status_body(404, "NotFound", "no route"))
                kind, ns, _name, _sub = route
                body = self._body()
                if kind == "VariantAutoscaling":
                    # status subresource semantics: status in the create
                    # body is ignored (kube-apiserver strips it)
                    body.pop("status", None)
                    errs = validate(_VA_SCHEMA, body)
                    if errs:
                        return self._send(422, _status_body(
                            422, "Invalid",
                            f"VariantAutoscaling is invalid: {'; '.join(errs)}",
                        ))
                obj = serde.decode(kind, body)
                try:
                    created = outer.cluster.create(obj)
                except ConflictError as e:
                    return self._send(409, _status_body(
                        409, "AlreadyExists", str(e)
                    ))
                return self._send(201, serde.encode(created))

            def do_PUT(self):
                self._count("PUT")
                if self._maybe_fail():
                    return
                route = _match(urlparse(self.path).path)
                if route is None:
                    return self._send(404, _status_body(404, "NotFound", "no route"))
                kind, ns, name, _sub = route
                body = self._body()
                if kind == "VariantAutoscaling":
                    # status subresource semantics: PUT to the main
                    # resource cannot change status — replace the body's
                    # status with the stored one before validation/update
                    stored = outer.cluster.try_get(kind, ns or "", name)
                    if stored is not None:
                        stored_status = serde.encode(stored).get("status")
                        if stored_status is not None:
                            body["status"] = stored_status
                        else:
                            body.pop("status", None)
                    errs = validate(_VA_SCHEMA, body)
                    if errs:
                        return self._send(422, _status_body(
                            422, "Invalid",
                            f"VariantAutoscaling is invalid: {'; '.join(errs)}",
                        ))
                obj = serde.decode(kind, body)
                # API-server semantics: metadata.generation bumps when
                # the spec (not status/metadata) changes
                prev = outer.cluster.try_get(kind, ns or "", name)
                bump = False
                if prev is not None:
                    prev_spec = serde.encode(prev).get("spec")
                    if prev_spec != body.get("spec"):
                        bump = True
                try:
                    updated = outer.cluster.update(obj, bump_generation=bump)
                except NotFoundError:
                    return self._send(404, _status_body(404, "NotFound", "not found"))
                except ConflictError as e:
                    # optimistic-concurrency loss → kube-shaped 409
                    return self._send(409, _status_body(409, "Conflict", str(e)))
                return self._send(200, serde.encode(updated))

            def do_PATCH(self):
                self._count("PATCH")
                if self._maybe_fail():
                    return
                route = _match(urlparse(self.path).path)
                if route is None:
                    return self._send(404, _status_body(404, "NotFound", "no route"))
                kind, ns, name, sub = route
                body = self._body()
                if sub == "scale":
                    replicas = int(
                        ((body.get("spec") or {}).get("replicas", 0))
                    )
                    try:
                        obj = outer.cluster.scale(kind, ns or "", name, replicas)
                    except NotFoundError:
                        return self._send(404, _status_body(
                            404, "NotFound", "not found"
                        ))
                    return self._send(200, serde.encode(obj))
                if sub == "status":
                    # RFC 7386 merge patch against the STORED object, then
                    # CRD validation of the RESULT — the #731 interaction:
                    # a partial desiredOptimizedAlloc patch merged into a
                    # VA with no prior alloc yields an object missing
                    # required fields → 422.
                    stored = outer.cluster.try_get(kind, ns or "", name)
                    if stored is None:
                        return self._send(404, _status_body(
                            404, "NotFound", "not found"
                        ))
                    stored_enc = serde.encode(stored)
                    merged = dict(stored_enc)
                    merged["status"] = merge_patch(
                        stored_enc.get("status") or {}, body.get("status") or {}
                    )
                    if kind == "VariantAutoscaling":
                        errs = validate(_VA_SCHEMA, merged)
                        if errs:
                            return self._send(422, _status_body(
                                422, "Invalid",
                                "VariantAutoscaling is invalid: "
                                + "; ".join(errs),
                            ))
                    obj = serde.decode(kind, merged)
                    try:
                        updated = outer.cluster.update_status(obj)
                    except NotFoundError:
                        return self._send(404, _status_body(
                            404, "NotFound", "not found"
                        ))
                    return self._send(200, serde.encode(updated))
                return self._send(400, _status_body(400, "BadRequest",
                                                    "unsupported patch"))

            def do_DELETE(self):
                self._count("DELETE")
                if self._maybe_fail():
                    return
                route = _match(urlparse(self.path).path)
                if route is None:
                    return self._send(404, _status_body(404, "NotFound", "no route"))
                kind, ns, name, _sub = route
                try:
                    outer.cluster.delete(kind, ns or "", name)
                except NotFoundError:
                    return self._send(404, _status_body(404, "NotFound", "not found"))
                return self._send(200, {"status": "Success"})

        self._closing = threading.Event()
        self.server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self._thread: Optional[threading.Thread] = None

    # --- fault injection / test hooks ---

    def fail_next(self, n: int = 1, code: int = 500) -> None:
        """Fail the next n mutating requests (POST/PUT/PATCH/DELETE)."""
        with self._fail_lock:
            self._fail_remaining = n
            self._fail_code = code

    def active_watch_count(self) -> int:
        with self._watch_lock:
            return len(self._watch_drops)

    def drop_watches(self) -> int:
        """Force-close every active watch stream; returns how many were
        dropped. Clients must transparently reconnect from their last
        observed resourceVersion without losing events."""
        with self._watch_lock:
            drops = list(self._watch_drops)
        for d in drops:
            d.set()
        return len(drops)

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
        self.drop_watches()
        self.server.shutdown()
        self.server.server_close()
        if self._thread:
            self._thread.join(timeout=5)
