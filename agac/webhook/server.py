"""Admission webhook HTTP(S) server.

Plain-stdlib HTTP server, mirroring the reference's plain ``net/http``
server (``pkg/webhoook/webhook.go:14-85``): ``GET /healthz`` and
``POST /validate-endpointgroupbinding``; TLS when cert+key files are given.
"""

from __future__ import annotations

import json
import logging
import ssl
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional, Tuple

from . import validator

logger = logging.getLogger(__name__)


class WebhookHandler(BaseHTTPRequestHandler):
    disable_nagle_algorithm = True
    # quiet default request logging into our logger
    def log_message(self, fmt, *args):  # noqa: A003
        logger.debug(fmt, *args)

    def _write(self, code: int, body: bytes, content_type: str = "application/json"):
        self.send_response(code)
        self.send_header("Content-Type", content_type)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def do_GET(self):  # noqa: N802
        if self.path == "/healthz":
            logger.debug("healthz")
            self._write(200, b"", "text/plain")
        else:
            self._write(404, b"not found", "text/plain")

    def do_POST(self):  # noqa: N802
        if self.path != "/validate-endpointgroupbinding":
            self._write(404, b"not found", "text/plain")
            return
        review, err = self._parse_request()
        if err is not None:
            self._write(400, err.encode(), "text/plain")
            return
        response = validator.validate(review)
        from .. import metrics

        allowed = ((response.get("response") or {}).get("allowed", False))
        metrics.observe_webhook_review(
            (review.get("request") or {}).get("operation", ""),
            "allowed" if allowed else "denied",
        )
        self._write(200, json.dumps(response).encode())

    def _parse_request(self) -> Tuple[Optional[dict], Optional[str]]:
        """Content-type + AdmissionReview decoding
        (reference pkg/webhoook/webhook.go:61-85)."""
        if self.headers.get("Content-Type") != "application/json":
            return None, "invalid Content-Type"
        length = int(self.headers.get("Content-Length") or 0)
        body = self.rfile.read(length) if length else b""
        if not body:
            return None, "empty body"
        try:
            review = json.loads(body)
        except ValueError as e:
            return None, f"failed to unmarshal body: {e}"
        if not isinstance(review, dict) or review.get("request") is None:
            return None, "empty request"
        return review, None


class WebhookServer:
    """Owns the listening socket; ``serve_forever`` in the caller's thread
    (CLI) or via ``start()`` for tests."""

    def __init__(
        self,
        port: int,
        tls_cert_file: str = "",
        tls_key_file: str = "",
        host: str = "",
    ):
        self.httpd = ThreadingHTTPServer((host, port), WebhookHandler)
        self.ssl_enabled = bool(tls_cert_file and tls_key_file)
        if self.ssl_enabled:
            context = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
            context.load_cert_chain(tls_cert_file, tls_key_file)
            self.httpd.socket = context.wrap_socket(self.httpd.socket, server_side=True)
        logger.info(
            "Listening on :%d, SSL is %s", self.port, str(self.ssl_enabled).lower()
        )

    @property
    def port(self) -> int:
        return self.httpd.server_address[1]

    def serve_forever(self):
        self.httpd.serve_forever()

    def start(self) -> threading.Thread:
        thread = threading.Thread(
            target=self.httpd.serve_forever, name="webhook-server", daemon=True
        )
        thread.start()
        return thread

    def shutdown(self):
        self.httpd.shutdown()
        self.httpd.server_close()


def serve(port: int, tls_cert_file: str = "", tls_key_file: str = ""):
    """Blocking entry point (reference ``Server``, webhook.go:14-33)."""
    WebhookServer(port, tls_cert_file, tls_key_file).serve_forever()
