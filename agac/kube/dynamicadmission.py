"""Dynamic admission: the apiserver side of ValidatingWebhookConfiguration
consumed AS API OBJECTS.

Real clusters activate admission by ``kubectl apply``ing a
ValidatingWebhookConfiguration; the apiserver then POSTs AdmissionReview
to the webhook's clientConfig (service reference or URL) on matching
writes, honoring failurePolicy and the caBundle.  This module gives the
hermetic apiserver the same mechanism, so
``agac apply -f config/webhook/manifests.yaml`` turns admission on exactly
like the reference's kind e2e does against a real cluster
(e2e/e2e_test.go:77-103).

Service references need name resolution (clusters use
``<name>.<namespace>.svc``); the store exposes an injectable
``webhook_service_resolver(service_ref) -> base_url`` — the hermetic e2e
points it at the running ``agac webhook`` server; unresolvable services
fall back to failurePolicy like an unreachable webhook would.
"""

from __future__ import annotations

import base64
import hashlib
import logging
import os
import tempfile
import uuid
from typing import Optional

from .admission import AdmissionDeniedError

logger = logging.getLogger(__name__)

_ca_files: dict = {}  # sha256(pem) -> temp file path


def _ca_bundle_file(ca_bundle_b64: str) -> Optional[str]:
    """Materialize a caBundle (base64 PEM) as a file for requests' verify=."""
    try:
        pem = base64.b64decode(ca_bundle_b64)
    except Exception:
        return None
    digest = hashlib.sha256(pem).hexdigest()
    path = _ca_files.get(digest)
    if path is None or not os.path.exists(path):
        fd, path = tempfile.mkstemp(prefix="agac-ca-", suffix=".pem")
        with os.fdopen(fd, "wb") as f:
            f.write(pem)
        _ca_files[digest] = path
    return path


def _rule_matches(rule, gvr, operation: str, subresource=None) -> bool:
    ops = rule.operations or []
    if "*" not in ops and operation not in ops:
        return False
    groups = rule.api_groups or []
    if "*" not in groups and gvr.group not in groups:
        return False
    resources = rule.resources or []
    if subresource:
        # a subresource write matches only "<plural>/<sub>" or "*/<sub>"
        return (
            f"{gvr.plural}/{subresource}" in resources
            or f"*/{subresource}" in resources
        )
    return "*" in resources or gvr.plural in resources


def _call_webhook(webhook, url: str, verify, kind: str, operation: str,
                  old, new):
    import requests

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
    response = requests.post(
        url, json=review, headers={"Content-Type": "application/json"},
        timeout=float(webhook.timeout_seconds or 10), verify=verify,
    )
    response.raise_for_status()
    result = (response.json() or {}).get("response") or {}
    if not result.get("allowed", False):
        message = ((result.get("status") or {}).get("message")) or "denied"
        raise AdmissionDeniedError(
            f"admission webhook {webhook.name!r} denied the request: {message}"
        )


def admit(store, kind: str, operation: str, old, new, subresource=None):
    """Evaluate every registered ValidatingWebhookConfiguration against
    this write.  Raises AdmissionDeniedError to veto."""
    from . import k8swire

    gvr = k8swire.BY_KIND.get(kind)
    if gvr is None or kind == "ValidatingWebhookConfiguration":
        return
    try:
        configs, _ = store.list("ValidatingWebhookConfiguration")
    except Exception:
        return
    for config in configs:
        for webhook in config.webhooks:
            if not any(
                _rule_matches(r, gvr, operation, subresource)
                for r in webhook.rules
            ):
                continue
            cc = webhook.client_config
            url = cc.url
            verify = True
            if url is None and cc.service is not None:
                resolver = getattr(store, "webhook_service_resolver", None)
                base = resolver(cc.service) if resolver else None
                if base:
                    url = base.rstrip("/") + (cc.service.path or "")
            if cc.ca_bundle:
                ca = _ca_bundle_file(cc.ca_bundle)
                if ca:
                    verify = ca
            try:
                if url is None:
                    raise ConnectionError(
                        f"webhook service "
                        f"{cc.service.namespace}/{cc.service.name} unresolvable"
                        if cc.service else "webhook has no url or service"
                    )
                _call_webhook(webhook, url, verify, kind, operation, old, new)
            except AdmissionDeniedError:
                raise
            except Exception as e:
                if (webhook.failure_policy or "Fail") == "Ignore":
                    logger.warning(
                        "webhook %s failed (%s); failurePolicy=Ignore admits",
                        webhook.name, e,
                    )
                    continue
                raise AdmissionDeniedError(
                    f"failed calling webhook {webhook.name!r} "
                    f"(failurePolicy=Fail): {e}"
                ) from e
