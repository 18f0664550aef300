"""EndpointGroupBinding admission validator.

Behavior parity with reference ``pkg/webhoook/endpointgroupbinding/
validator.go:15-76``: only UPDATE operations are validated; changing
``spec.endpointGroupArn`` is rejected with 403 "Spec.EndpointGroupArn is
immutable"; CREATE/DELETE and other kinds pass through (unknown kinds are
rejected with 400).
"""

from __future__ import annotations

import logging

from ..apis.endpointgroupbinding import EndpointGroupBinding
from ..apis.meta import from_dict

logger = logging.getLogger(__name__)


def review_response(uid: str, allowed: bool, code: int, reason: str) -> dict:
    return {
        "kind": "AdmissionReview",
        "apiVersion": "admission.k8s.io/v1",
        "response": {
            "uid": uid,
            "allowed": allowed,
            "status": {"code": code, "message": reason},
        },
    }


def validate(review: dict) -> dict:
    """Takes a decoded AdmissionReview dict, returns the response review."""
    request = review.get("request") or {}
    uid = request.get("uid", "")

    kind = (request.get("kind") or {}).get("kind")
    if kind != "EndpointGroupBinding":
        message = f"{kind} is not supported"
        logger.error(message)
        return review_response(uid, False, 400, message)

    if request.get("operation") != "UPDATE":
        logger.debug("Operation is not Update")
        return review_response(uid, True, 200, "")

    old_raw = request.get("oldObject")
    if old_raw is None:
        logger.debug("OldObject is nil")
        return review_response(uid, True, 200, "")

    new_raw = request.get("object")
    try:
        previous = from_dict(EndpointGroupBinding, old_raw)
        new = from_dict(EndpointGroupBinding, new_raw or {})
    except Exception as e:
        logger.error("failed to decode objects: %s", e)
        return review_response(uid, False, 500, str(e))

    if previous.spec.endpoint_group_arn != new.spec.endpoint_group_arn:
        message = "Spec.EndpointGroupArn is immutable"
        logger.error(message)
        return review_response(uid, False, 403, message)

    return review_response(uid, True, 200, "valid")
