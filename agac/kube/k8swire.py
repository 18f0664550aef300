"""Kubernetes wire-format registry: kind ↔ REST path mapping, list kinds
and Status error objects.

This is what lets agac speak to (and serve) the *real* Kubernetes REST
surface: ``/api/v1/namespaces/{ns}/services``,
``/apis/networking.k8s.io/v1/.../ingresses``,
``/apis/operator.h3poteto.dev/v1alpha1/.../endpointgroupbindings`` (the CRD
the reference registers via its generated clientset), watch via
``?watch=true`` streaming, and k8s ``Status`` failure objects.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from .store import (
    AlreadyExistsError,
    APIError,
    ConflictError,
    GoneError,
    NotFoundError,
)


@dataclass(frozen=True)
class GroupVersionResource:
    group: str  # "" for core
    version: str
    plural: str
    kind: str
    namespaced: bool = True

    @property
    def api_prefix(self) -> str:
        if self.group == "":
            return f"/api/{self.version}"
        return f"/apis/{self.group}/{self.version}"

    def path(self, namespace: Optional[str] = None, name: Optional[str] = None,
             subresource: Optional[str] = None) -> str:
        segments = [self.api_prefix.strip("/")]
        if namespace:
            segments += ["namespaces", namespace]
        segments.append(self.plural)
        if name:
            segments.append(name)
        if subresource:
            segments.append(subresource)
        return "/" + "/".join(segments)

    @property
    def api_version(self) -> str:
        return self.version if self.group == "" else f"{self.group}/{self.version}"


GVRS = [
    GroupVersionResource("", "v1", "services", "Service"),
    GroupVersionResource("", "v1", "events", "Event"),
    GroupVersionResource("coordination.k8s.io", "v1", "leases", "Lease"),
    GroupVersionResource("networking.k8s.io", "v1", "ingresses", "Ingress"),
    GroupVersionResource(
        "operator.h3poteto.dev", "v1alpha1", "endpointgroupbindings", "EndpointGroupBinding"
    ),
    GroupVersionResource(
        "admissionregistration.k8s.io", "v1", "validatingwebhookconfigurations",
        "ValidatingWebhookConfiguration", namespaced=False,
    ),
]

BY_KIND = {g.kind: g for g in GVRS}
BY_PLURAL = {(g.group, g.version, g.plural): g for g in GVRS}


def gvr_for_kind(kind: str) -> GroupVersionResource:
    return BY_KIND[kind]


def resolve_path(parts):
    """Parse a k8s-style URL path into (gvr, namespace, name, subresource)
    or None if the path is not a known k8s resource route.

    Accepted shapes (after splitting on '/'):
      api/v1/<plural>[...]                         (all-namespaces)
      api/v1/namespaces/<ns>/<plural>[/<name>[/status]]
      apis/<group>/<version>/<plural>[...]
      apis/<group>/<version>/namespaces/<ns>/<plural>[/<name>[/status]]
    """
    if not parts:
        return None
    if parts[0] == "api" and len(parts) >= 2:
        group, version, rest = "", parts[1], parts[2:]
    elif parts[0] == "apis" and len(parts) >= 3:
        group, version, rest = parts[1], parts[2], parts[3:]
    else:
        return None

    namespace = None
    if len(rest) >= 2 and rest[0] == "namespaces":
        namespace = rest[1]
        rest = rest[2:]
    if not rest:
        return None
    plural, rest = rest[0], rest[1:]
    gvr = BY_PLURAL.get((group, version, plural))
    if gvr is None:
        return None
    name = rest[0] if rest else None
    subresource = rest[1] if len(rest) > 1 else None
    if len(rest) > 2:
        return None
    return gvr, namespace, name, subresource


_ERROR_REASONS = {
    NotFoundError: "NotFound",
    AlreadyExistsError: "AlreadyExists",
    ConflictError: "Conflict",
    GoneError: "Expired",
}


def status_for_error(e: APIError) -> dict:
    """k8s metav1.Status failure object."""
    from .admission import AdmissionDeniedError
    from .validation import ValidationError

    reason = _ERROR_REASONS.get(type(e))
    if reason is None and isinstance(e, AdmissionDeniedError):
        reason = "Forbidden"
    if reason is None and isinstance(e, ValidationError):
        reason = "Invalid"
    return {
        "kind": "Status",
        "apiVersion": "v1",
        "status": "Failure",
        "message": str(e),
        "reason": reason or "InternalError",
        "code": e.code,
    }


def error_for_status(status: dict, http_code: int) -> APIError:
    from .admission import AdmissionDeniedError

    reason = status.get("reason", "")
    message = status.get("message", "")
    from .validation import ValidationError

    mapping = {
        "NotFound": NotFoundError,
        "AlreadyExists": AlreadyExistsError,
        "Conflict": ConflictError,
        "Expired": GoneError,
        "Gone": GoneError,
        "Forbidden": AdmissionDeniedError,
        "Invalid": ValidationError,
    }
    cls = mapping.get(reason)
    if cls is not None:
        return cls(message)
    return APIError(message or f"HTTP {http_code}", http_code)
