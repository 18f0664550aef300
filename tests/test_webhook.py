"""Webhook server + validator tests (reference pkg/webhoook/webhook_test.go,
218 LoC httptest suite): healthz, weight-change allowed, ARN-change
rejected, malformed request handling — over a real HTTP socket."""

import json
import urllib.error
import urllib.request

import pytest

from agac.apis.meta import to_dict
from agac.fixture import endpoint_group_binding
from agac.webhook.server import WebhookServer
from agac.webhook.validator import validate


def admission_review(old, new, operation="UPDATE", kind="EndpointGroupBinding"):
    return {
        "kind": "AdmissionReview",
        "apiVersion": "admission.k8s.io/v1",
        "request": {
            "uid": "test-uid-123",
            "kind": {"group": "operator.h3poteto.dev", "version": "v1alpha1", "kind": kind},
            "operation": operation,
            "object": to_dict(new) if new is not None else None,
            "oldObject": to_dict(old) if old is not None else None,
        },
    }


class TestValidator:
    def test_update_weight_allowed(self):
        old = endpoint_group_binding(weight=100)
        new = endpoint_group_binding(weight=200)
        response = validate(admission_review(old, new))
        assert response["response"]["allowed"] is True
        assert response["response"]["uid"] == "test-uid-123"

    def test_update_arn_rejected(self):
        old = endpoint_group_binding()
        new = endpoint_group_binding(endpoint_group_arn="arn:aws:globalaccelerator::1:other")
        response = validate(admission_review(old, new))
        assert response["response"]["allowed"] is False
        assert response["response"]["status"]["code"] == 403
        assert "immutable" in response["response"]["status"]["message"]

    def test_create_allowed(self):
        new = endpoint_group_binding()
        response = validate(admission_review(None, new, operation="CREATE"))
        assert response["response"]["allowed"] is True

    def test_update_without_old_object_allowed(self):
        new = endpoint_group_binding()
        response = validate(admission_review(None, new))
        assert response["response"]["allowed"] is True

    def test_unsupported_kind_rejected(self):
        response = validate(admission_review(None, None, kind="Deployment"))
        assert response["response"]["allowed"] is False
        assert response["response"]["status"]["code"] == 400


@pytest.fixture(scope="module")
def server():
    s = WebhookServer(port=0)  # ephemeral port, no TLS
    s.start()
    yield s
    s.shutdown()


def post(server, path, body: bytes, content_type="application/json"):
    req = urllib.request.Request(
        f"http://127.0.0.1:{server.port}{path}",
        data=body,
        headers={"Content-Type": content_type},
        method="POST",
    )
    return urllib.request.urlopen(req, timeout=5)


class TestServerHTTP:
    def test_healthz(self, server):
        with urllib.request.urlopen(
            f"http://127.0.0.1:{server.port}/healthz", timeout=5
        ) as resp:
            assert resp.status == 200

    def test_validate_allowed_roundtrip(self, server):
        old = endpoint_group_binding(weight=1)
        new = endpoint_group_binding(weight=2)
        with post(server, "/validate-endpointgroupbinding",
                  json.dumps(admission_review(old, new)).encode()) as resp:
            body = json.loads(resp.read())
        assert body["response"]["allowed"] is True
        assert body["apiVersion"] == "admission.k8s.io/v1"

    def test_validate_rejected_roundtrip(self, server):
        old = endpoint_group_binding()
        new = endpoint_group_binding(endpoint_group_arn="arn:changed")
        with post(server, "/validate-endpointgroupbinding",
                  json.dumps(admission_review(old, new)).encode()) as resp:
            body = json.loads(resp.read())
        assert body["response"]["allowed"] is False
        assert body["response"]["status"]["code"] == 403

    def test_wrong_content_type_400(self, server):
        with pytest.raises(urllib.error.HTTPError) as exc:
            post(server, "/validate-endpointgroupbinding", b"{}", content_type="text/plain")
        assert exc.value.code == 400

    def test_empty_body_400(self, server):
        with pytest.raises(urllib.error.HTTPError) as exc:
            post(server, "/validate-endpointgroupbinding", b"")
        assert exc.value.code == 400

    def test_invalid_json_400(self, server):
        with pytest.raises(urllib.error.HTTPError) as exc:
            post(server, "/validate-endpointgroupbinding", b"not json")
        assert exc.value.code == 400

    def test_missing_request_400(self, server):
        with pytest.raises(urllib.error.HTTPError) as exc:
            post(server, "/validate-endpointgroupbinding", b'{"kind": "AdmissionReview"}')
        assert exc.value.code == 400

    def test_unknown_path_404(self, server):
        with pytest.raises(urllib.error.HTTPError) as exc:
            post(server, "/nope", b"{}")
        assert exc.value.code == 404


class TestOperationScope:
    def test_delete_and_connect_ops_allowed(self):
        """Reference validator.go:22-26: any non-UPDATE operation passes
        (DELETE, CONNECT, CREATE) — the webhook only guards mutation of
        the ARN."""
        for op in ("DELETE", "CONNECT", "CREATE"):
            review = admission_review(None, endpoint_group_binding(), operation=op)
            response = validate(review)
            assert response["response"]["allowed"] is True, op
