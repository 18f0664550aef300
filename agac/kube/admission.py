"""Admission plumbing for the API store.

The kube-apiserver side of ValidatingWebhookConfiguration
(``config/webhook/manifests.yaml``): registered webhooks are consulted on
matching operations before a write is committed; a disallowed review fails
the request with 403, exactly what the reference's kind e2e asserts
(``e2e/e2e_test.go:77-103`` — ARN update rejected, weight update allowed).

Two transports:
- ``LocalAdmission``  — calls a validator function in-process;
- ``HTTPAdmission``   — POSTs a real AdmissionReview to a webhook server
  over HTTP(S) (the production shape; used by the hermetic e2e suite
  against ``agac.webhook.server``).
"""

from __future__ import annotations

import logging
import uuid
from typing import Callable, List, Optional

from .store import APIError

logger = logging.getLogger(__name__)


class AdmissionDeniedError(APIError):
    def __init__(self, message: str = "admission webhook denied the request"):
        super().__init__(message, 403)


class AdmissionWebhook:
    """One registered webhook: kind + operations filter + a transport."""

    def __init__(self, kinds: List[str], operations: List[str], review_fn: Callable[[dict], dict]):
        self.kinds = set(kinds)
        self.operations = set(operations)
        self.review_fn = review_fn

    def admit(self, kind: str, operation: str, old: Optional[dict], new: Optional[dict]):
        """Raises AdmissionDeniedError if the webhook disallows the write."""
        if kind not in self.kinds or operation not in self.operations:
            return
        review = {
            "kind": "AdmissionReview",
            "apiVersion": "admission.k8s.io/v1",
            "request": {
                "uid": str(uuid.uuid4()),
                "kind": {"kind": kind},
                "operation": operation,
                "object": new,
                "oldObject": old,
            },
        }
        response = self.review_fn(review)
        result = (response or {}).get("response") or {}
        if not result.get("allowed", False):
            message = ((result.get("status") or {}).get("message")) or "denied"
            raise AdmissionDeniedError(
                f"admission webhook denied the request: {message}"
            )


def local_admission(kinds: List[str], operations: List[str], validate_fn) -> AdmissionWebhook:
    """In-process transport (validate_fn: AdmissionReview dict -> response)."""
    return AdmissionWebhook(kinds, operations, validate_fn)


def http_admission(
    kinds: List[str],
    operations: List[str],
    url: str,
    timeout: float = 10.0,
    verify=True,
    failure_policy: str = "Fail",
) -> AdmissionWebhook:
    """HTTP transport: POST the AdmissionReview to ``url``.

    ``failure_policy`` mirrors ValidatingWebhookConfiguration: when the
    webhook is unreachable/broken, ``Fail`` (the k8s default) rejects the
    write, ``Ignore`` admits it."""
    import requests

    def review_fn(review: dict) -> dict:
        try:
            response = requests.post(
                url,
                json=review,
                headers={"Content-Type": "application/json"},
                timeout=timeout,
                verify=verify,
            )
            response.raise_for_status()
            return response.json()
        except Exception as e:
            if failure_policy == "Ignore":
                logger.warning(
                    "admission webhook %s unreachable (%s); failurePolicy="
                    "Ignore admits the request", url, e,
                )
                return {"response": {"allowed": True}}
            raise AdmissionDeniedError(
                f"failed calling webhook {url} (failurePolicy=Fail): {e}"
            ) from e

    return AdmissionWebhook(kinds, operations, review_fn)
