"""Controller health endpoints.

The reference controller process exposes NO health surface (only the
webhook has /healthz) — deployments can't probe it.  ``serve_health``
runs a tiny HTTP listener with:

- ``/healthz`` — process liveness (always 200 while the thread runs);
- ``/readyz``  — 200 once every informer cache has synced (Manager
  .is_ready), 503 before that or when no manager is active (e.g. this
  replica is a leader-election standby, which is a correct NotReady for a
  readiness gate that routes traffic, and harmless for plain liveness).

Wired into ``agac controller --health-port`` and the Helm chart's probes.
"""

from __future__ import annotations

import json
import logging
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

logger = logging.getLogger(__name__)


class _Handler(BaseHTTPRequestHandler):
    ready_fn = staticmethod(lambda: False)

    def log_message(self, fmt, *args):  # noqa: A003
        logger.debug(fmt, *args)

    def do_GET(self):  # noqa: N802
        path = self.path.split("?", 1)[0]
        if path == "/healthz":
            self._respond(200, {"status": "ok"})
        elif path == "/readyz":
            if self.ready_fn():
                self._respond(200, {"status": "ready"})
            else:
                self._respond(503, {"status": "not ready"})
        else:
            self._respond(404, {"status": "not found"})

    def _respond(self, code: int, payload: dict):
        body = json.dumps(payload).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)


class HealthServer:
    def __init__(self, port: int, ready_fn, host: str = ""):
        handler = type("BoundHealth", (_Handler,), {"ready_fn": staticmethod(ready_fn)})
        self.httpd = ThreadingHTTPServer((host, port), handler)
        self.httpd.daemon_threads = True

    @property
    def port(self) -> int:
        return self.httpd.server_address[1]

    def start(self):
        thread = threading.Thread(
            target=self.httpd.serve_forever, name="agac-health", daemon=True
        )
        thread.start()
        return thread

    def shutdown(self):
        self.httpd.shutdown()
        self.httpd.server_close()
