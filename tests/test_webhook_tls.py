"""Webhook TLS path: serve with a self-signed cert and validate over HTTPS
(the reference's production shape — cert-manager-issued certs mounted into
the webhook pod)."""

import json
import ssl
import subprocess
import urllib.request

import pytest

from agac.apis.meta import to_dict
from agac.fixture import endpoint_group_binding
from agac.webhook.server import WebhookServer


@pytest.fixture(scope="module")
def tls_files(tmp_path_factory):
    d = tmp_path_factory.mktemp("certs")
    cert, key = d / "tls.crt", d / "tls.key"
    subprocess.run(
        [
            "openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
            "-keyout", str(key), "-out", str(cert), "-days", "1",
            "-subj", "/CN=127.0.0.1",
            "-addext", "subjectAltName=IP:127.0.0.1",
        ],
        check=True,
        capture_output=True,
    )
    return str(cert), str(key)


@pytest.fixture(scope="module")
def tls_server(tls_files):
    cert, key = tls_files
    server = WebhookServer(port=0, tls_cert_file=cert, tls_key_file=key)
    assert server.ssl_enabled
    server.start()
    yield server, cert
    server.shutdown()


def test_validate_over_https(tls_server):
    server, cert = tls_server
    context = ssl.create_default_context(cafile=cert)
    old = endpoint_group_binding()
    new = endpoint_group_binding(endpoint_group_arn="arn:changed")
    review = {
        "kind": "AdmissionReview",
        "apiVersion": "admission.k8s.io/v1",
        "request": {
            "uid": "tls-uid",
            "kind": {"kind": "EndpointGroupBinding"},
            "operation": "UPDATE",
            "object": to_dict(new),
            "oldObject": to_dict(old),
        },
    }
    req = urllib.request.Request(
        f"https://127.0.0.1:{server.port}/validate-endpointgroupbinding",
        data=json.dumps(review).encode(),
        headers={"Content-Type": "application/json"},
        method="POST",
    )
    with urllib.request.urlopen(req, timeout=5, context=context) as resp:
        body = json.loads(resp.read())
    assert body["response"]["allowed"] is False
    assert body["response"]["status"]["code"] == 403


def test_healthz_over_https(tls_server):
    server, cert = tls_server
    context = ssl.create_default_context(cafile=cert)
    with urllib.request.urlopen(
        f"https://127.0.0.1:{server.port}/healthz", timeout=5, context=context
    ) as resp:
        assert resp.status == 200
