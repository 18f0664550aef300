"""HTTP API server over the in-memory store.

The kube-apiserver stand-in for e2e testing with a real network boundary
(the analogue of the reference's kind-based e2e tier, ``e2e/``): serves the
``APIStore`` over HTTP with list/get/create/update/update-status/delete and
chunked-streaming watch, so ``RestKubeClient`` + the controller manager run
against it exactly as they would against a remote API server.

Wire scheme (generic, k8s-shaped):

- ``GET    /apis/{kind}[?namespace=ns]``        → {"items": [...], "resourceVersion": N}
- ``GET    /apis/{kind}/{ns}/{name}``           → object
- ``POST   /apis/{kind}``                        → created object
- ``PUT    /apis/{kind}/{ns}/{name}``           → updated object
- ``PUT    /apis/{kind}/{ns}/{name}/status``    → updated object
- ``DELETE /apis/{kind}/{ns}/{name}``           → {}
- ``GET    /watch/{kind}[?namespace=&resourceVersion=N]``
      → ndjson stream of {"type", "object", "resourceVersion"}
- ``GET    /healthz``

The REAL Kubernetes REST surface is served too (``_k8s_route``):
``/api/v1/namespaces/{ns}/services``, ``/apis/<group>/<version>/…``,
``?watch=true`` streaming with apiserver event framing, ``<Kind>List``
bodies and ``Status`` failure objects — so ``K8sKubeClient`` (and any
client-go-shaped client) runs against this server unchanged.

Errors: JSON ``{"code", "reason", "message"}`` with the matching HTTP
status; reasons NotFound / AlreadyExists / Conflict / Gone map back to the
typed store errors in the client.
"""

from __future__ import annotations

import json
import logging
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from urllib.parse import parse_qs, urlparse

from ..apis.meta import from_dict, to_dict
from .client import class_for_kind
from .store import (
    AlreadyExistsError,
    APIError,
    APIStore,
    ConflictError,
    GoneError,
    NotFoundError,
)

logger = logging.getLogger(__name__)


def _error_body(e: APIError) -> dict:
    from .admission import AdmissionDeniedError

    reason = {
        NotFoundError: "NotFound",
        AlreadyExistsError: "AlreadyExists",
        ConflictError: "Conflict",
        GoneError: "Gone",
        AdmissionDeniedError: "Forbidden",
    }.get(type(e), "InternalError")
    return {"code": e.code, "reason": reason, "message": str(e)}


class _Handler(BaseHTTPRequestHandler):
    protocol_version = "HTTP/1.1"
    disable_nagle_algorithm = True  # small JSON bodies; latency over batching
    store: APIStore = None  # set by server factory
    watch_idle_seconds = 5.0  # heartbeat/bookmark cadence on idle watches
    bearer_token = None  # require `Authorization: Bearer <token>` when set

    def _authorized(self) -> bool:
        """Bearer-token authn (apiserver static-token file shape); /healthz
        stays open like a real apiserver's healthz with anonymous-auth."""
        if self.bearer_token is None:
            return True
        if self.path.split("?", 1)[0] == "/healthz":
            return True
        header = self.headers.get("Authorization", "")
        if header == f"Bearer {self.bearer_token}":
            return True
        self._json(401, {
            "kind": "Status", "apiVersion": "v1", "status": "Failure",
            "reason": "Unauthorized", "code": 401,
            "message": "Unauthorized",
        })
        return False

    def log_message(self, fmt, *args):  # noqa: A003
        logger.debug(fmt, *args)

    # -- helpers -----------------------------------------------------------
    def _json(self, code: int, payload: dict):
        body = json.dumps(payload).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def _api_error(self, e: APIError):
        self._json(e.code, _error_body(e))

    def _obj_with_kind(self, obj) -> dict:
        d = to_dict(obj)
        d["kind"] = type(obj).kind
        d["apiVersion"] = type(obj).api_version
        return d

    def _read_body(self):
        length = int(self.headers.get("Content-Length") or 0)
        return json.loads(self.rfile.read(length)) if length else {}

    def _route(self):
        parsed = urlparse(self.path)
        parts = [p for p in parsed.path.split("/") if p]
        query = {k: v[0] for k, v in parse_qs(parsed.query).items()}
        return parts, query

    # -- k8s-compatible routes ---------------------------------------------
    def _k8s_route(self, parts, query, method: str) -> bool:
        """Serve real Kubernetes REST paths (``/api/v1/namespaces/...``,
        ``/apis/<group>/<version>/...``, ``?watch=true``) so a stock
        client-go-shaped client can talk to this server.  Returns True if
        the path was a k8s route (handled), False to fall through."""
        from . import k8swire

        resolved = k8swire.resolve_path(parts)
        if resolved is None:
            return False
        gvr, namespace, name, subresource = resolved
        try:
            if method == "GET" and name is None:
                if query.get("watch") in ("true", "1"):
                    rv = query.get("resourceVersion")
                    self._serve_watch_stream(
                        gvr.kind,
                        namespace,
                        int(rv) if rv else None,
                        k8s_style=True,
                        bookmarks=query.get("allowWatchBookmarks") in ("true", "1"),
                        timeout_seconds=float(query["timeoutSeconds"])
                        if query.get("timeoutSeconds") else None,
                    )
                    return True
                limit = int(query["limit"]) if query.get("limit") else None
                items, rv, next_token = self.store.list_page(
                    gvr.kind, namespace, limit, query.get("continue") or None
                )
                metadata = {"resourceVersion": str(rv)}
                if next_token:
                    metadata["continue"] = next_token
                self._json(
                    200,
                    {
                        "kind": f"{gvr.kind}List",
                        "apiVersion": gvr.api_version,
                        "metadata": metadata,
                        "items": [self._obj_with_kind(o) for o in items],
                    },
                )
            elif method == "GET":
                obj = self.store.get(gvr.kind, namespace or "", name)
                self._json(200, self._obj_with_kind(obj))
            elif method == "POST" and name is None:
                cls = class_for_kind(gvr.kind)
                obj = from_dict(cls, self._read_body())
                if namespace:
                    obj.metadata.namespace = namespace
                created = self.store.create(obj)
                self._json(201, self._obj_with_kind(created))
            elif method == "PUT" and name is not None:
                cls = class_for_kind(gvr.kind)
                obj = from_dict(cls, self._read_body())
                obj.metadata.namespace = namespace or obj.metadata.namespace
                obj.metadata.name = name
                if subresource == "status":
                    updated = self.store.update_status(obj)
                elif subresource is None:
                    updated = self.store.update(obj)
                else:
                    self._json(404, k8swire.status_for_error(NotFoundError("unknown subresource")))
                    return True
                self._json(200, self._obj_with_kind(updated))
            elif method == "PATCH" and name is not None:
                updated = self.store.patch(
                    gvr.kind, namespace or "", name, self._read_body(), subresource
                )
                self._json(200, self._obj_with_kind(updated))
            elif method == "DELETE" and name is not None:
                self.store.delete(gvr.kind, namespace or "", name)
                self._json(
                    200,
                    {"kind": "Status", "apiVersion": "v1", "status": "Success"},
                )
            else:
                self._json(405, k8swire.status_for_error(APIError("method not allowed", 405)))
        except APIError as e:
            self._json(e.code, k8swire.status_for_error(e))
        except BrokenPipeError:
            pass
        return True

    # -- verbs -------------------------------------------------------------
    def do_GET(self):  # noqa: N802
        if not self._authorized():
            return
        parts, query = self._route()
        if self._k8s_route(parts, query, "GET"):
            return
        try:
            if parts == ["healthz"]:
                self._json(200, {"status": "ok"})
            elif len(parts) == 2 and parts[0] == "apis":
                limit = int(query["limit"]) if query.get("limit") else None
                items, rv, next_token = self.store.list_page(
                    parts[1], query.get("namespace"), limit, query.get("continue") or None
                )
                body = {
                    "items": [self._obj_with_kind(o) for o in items],
                    "resourceVersion": rv,
                }
                if next_token:
                    body["continue"] = next_token
                self._json(200, body)
            elif len(parts) == 4 and parts[0] == "apis":
                obj = self.store.get(parts[1], parts[2], parts[3])
                self._json(200, self._obj_with_kind(obj))
            elif len(parts) == 2 and parts[0] == "watch":
                self._serve_watch(parts[1], query)
            else:
                self._json(404, {"code": 404, "reason": "NotFound", "message": "no such route"})
        except APIError as e:
            self._api_error(e)
        except BrokenPipeError:
            pass

    def do_POST(self):  # noqa: N802
        if not self._authorized():
            return
        parts, query = self._route()
        if self._k8s_route(parts, query, "POST"):
            return
        try:
            if len(parts) == 2 and parts[0] == "apis":
                cls = class_for_kind(parts[1])
                obj = from_dict(cls, self._read_body())
                created = self.store.create(obj)
                self._json(201, self._obj_with_kind(created))
            else:
                self._json(404, {"code": 404, "reason": "NotFound", "message": "no such route"})
        except APIError as e:
            self._api_error(e)

    def do_PUT(self):  # noqa: N802
        if not self._authorized():
            return
        parts, query = self._route()
        if self._k8s_route(parts, query, "PUT"):
            return
        try:
            if len(parts) in (4, 5) and parts[0] == "apis":
                kind, ns, name = parts[1], parts[2], parts[3]
                cls = class_for_kind(kind)
                obj = from_dict(cls, self._read_body())
                obj.metadata.namespace = ns
                obj.metadata.name = name
                if len(parts) == 5 and parts[4] == "status":
                    updated = self.store.update_status(obj)
                else:
                    updated = self.store.update(obj)
                self._json(200, self._obj_with_kind(updated))
            else:
                self._json(404, {"code": 404, "reason": "NotFound", "message": "no such route"})
        except APIError as e:
            self._api_error(e)

    def do_PATCH(self):  # noqa: N802
        if not self._authorized():
            return
        parts, query = self._route()
        if self._k8s_route(parts, query, "PATCH"):
            return
        try:
            if len(parts) in (4, 5) and parts[0] == "apis":
                kind, ns, name = parts[1], parts[2], parts[3]
                subresource = parts[4] if len(parts) == 5 else None
                updated = self.store.patch(kind, ns, name, self._read_body(), subresource)
                self._json(200, self._obj_with_kind(updated))
            else:
                self._json(404, {"code": 404, "reason": "NotFound", "message": "no such route"})
        except APIError as e:
            self._api_error(e)

    def do_DELETE(self):  # noqa: N802
        if not self._authorized():
            return
        parts, query = self._route()
        if self._k8s_route(parts, query, "DELETE"):
            return
        try:
            if len(parts) == 4 and parts[0] == "apis":
                self.store.delete(parts[1], parts[2], parts[3])
                self._json(200, {})
            else:
                self._json(404, {"code": 404, "reason": "NotFound", "message": "no such route"})
        except APIError as e:
            self._api_error(e)

    # -- watch streaming ----------------------------------------------------
    def _serve_watch(self, kind: str, query: dict):
        rv = query.get("resourceVersion")
        self._serve_watch_stream(
            kind, query.get("namespace"), int(rv) if rv is not None else None,
            k8s_style=False,
            bookmarks=query.get("allowWatchBookmarks") in ("true", "1"),
            timeout_seconds=float(query["timeoutSeconds"])
            if query.get("timeoutSeconds") else None,
        )

    def _serve_watch_stream(self, kind, namespace, rv, k8s_style: bool,
                            bookmarks: bool = False,
                            timeout_seconds=None):
        """Chunked ndjson event stream.  k8s_style frames events as the real
        apiserver does ({"type", "object"}); the native scheme adds a
        top-level resourceVersion.  With ``allowWatchBookmarks=true``, idle
        periods emit BOOKMARK events carrying only metadata.resourceVersion
        (the apiserver's watch-bookmark contract) so clients can advance
        their resume point without object traffic."""
        import time as _time

        watch = self.store.watch(kind, namespace, rv)
        deadline = (
            _time.monotonic() + timeout_seconds if timeout_seconds else None
        )
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.send_header("Transfer-Encoding", "chunked")
        self.end_headers()
        try:
            while True:
                if deadline is not None and _time.monotonic() >= deadline:
                    # ?timeoutSeconds= elapsed: end the stream like a real
                    # apiserver; the client re-watches from its resume rv
                    break
                # Snapshot the rv BEFORE waiting: anything issued after this
                # point is still in the watch queue, so bookmarking at this
                # rv can never skip an undelivered event.
                bookmark_rv = self.store.latest_rv()
                event = watch.get(timeout=self.watch_idle_seconds)
                if event is None:
                    if watch.closed:
                        # the store-side subscription ended — end the HTTP
                        # stream too so the client reconnects and resumes
                        # from its last delivered rv (real apiservers close
                        # watches periodically; clients must handle it)
                        break
                    if bookmarks:
                        payload = {
                            "type": "BOOKMARK",
                            "object": {
                                "kind": kind,
                                "metadata": {"resourceVersion": str(bookmark_rv)},
                            },
                        }
                        if not k8s_style:
                            payload["resourceVersion"] = bookmark_rv
                        self._write_chunk(json.dumps(payload).encode() + b"\n")
                    else:
                        # heartbeat keeps half-open connections detectable
                        self._write_chunk(b"")
                    continue
                payload = {
                    "type": event.type,
                    "object": self._obj_with_kind(event.obj),
                }
                if not k8s_style:
                    payload["resourceVersion"] = event.resource_version
                self._write_chunk(json.dumps(payload).encode() + b"\n")
            # clean end-of-stream: terminate the chunked body so the client
            # sees EOF promptly instead of waiting for a socket teardown
            self.wfile.write(b"0\r\n\r\n")
            self.wfile.flush()
            self.close_connection = True
        except (BrokenPipeError, ConnectionResetError, OSError):
            pass
        finally:
            watch.stop()

    def _write_chunk(self, data: bytes):
        if not data:
            # zero-length would terminate chunked encoding; send a newline
            data = b"\n"
        self.wfile.write(f"{len(data):x}\r\n".encode() + data + b"\r\n")
        self.wfile.flush()


class APIServer:
    """Owns the HTTP listener for one APIStore."""

    def __init__(self, store: APIStore, port: int = 0, host: str = "127.0.0.1",
                 watch_idle_seconds: float = 5.0, bearer_token=None,
                 tls_cert_file: str = "", tls_key_file: str = ""):
        handler = type(
            "BoundHandler",
            (_Handler,),
            {"store": store, "watch_idle_seconds": watch_idle_seconds,
             "bearer_token": bearer_token},
        )
        self.httpd = ThreadingHTTPServer((host, port), handler)
        self.httpd.daemon_threads = True
        self.ssl_enabled = bool(tls_cert_file and tls_key_file)
        if self.ssl_enabled:
            import ssl

            context = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
            context.load_cert_chain(tls_cert_file, tls_key_file)
            self.httpd.socket = context.wrap_socket(
                self.httpd.socket, server_side=True
            )
        self.store = store

    @property
    def port(self) -> int:
        return self.httpd.server_address[1]

    @property
    def url(self) -> str:
        host, port = self.httpd.server_address[:2]
        scheme = "https" if self.ssl_enabled else "http"
        return f"{scheme}://{host}:{port}"

    def start(self) -> threading.Thread:
        thread = threading.Thread(
            target=self.httpd.serve_forever, name="agac-apiserver", daemon=True
        )
        thread.start()
        return thread

    def shutdown(self):
        self.httpd.shutdown()
        self.httpd.server_close()


def serve_store(store: APIStore, port: int):
    """Blocking CLI entry point."""
    server = APIServer(store, port, host="")
    logger.info("API server listening on :%d", server.port)
    server.httpd.serve_forever()
