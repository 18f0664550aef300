"""Structural (CRD-schema) validation for API objects.

The kube-apiserver validates CRD objects against their openAPIV3Schema
before admission; this module is that layer for the in-process store,
enforcing the constraints the CRD yaml declares
(``config/crd/endpointgroupbindings.yaml`` / reference
``pkg/apis/endpointgroupbinding/v1alpha1/types.go`` kubebuilder markers):
``spec.endpointGroupArn`` required, ``weight`` nullable int32, refs with
required ``name``.  All kinds get basic metadata validation.
"""

from __future__ import annotations

from ..apis.endpointgroupbinding import EndpointGroupBinding
from .store import APIError

_INT32_MAX = 2**31 - 1


class ValidationError(APIError):
    """422 Unprocessable Entity, reason Invalid (k8s parity)."""

    def __init__(self, message: str):
        super().__init__(message, 422)


def validate_object(obj) -> None:
    """Raises ValidationError for schema violations; no-op otherwise."""
    if not obj.metadata.name:
        raise ValidationError(f"{type(obj).kind}: metadata.name is required")
    if isinstance(obj, EndpointGroupBinding):
        _validate_egb(obj)


def _validate_egb(obj: EndpointGroupBinding) -> None:
    if not obj.spec.endpoint_group_arn:
        raise ValidationError(
            "EndpointGroupBinding: spec.endpointGroupArn is required"
        )
    weight = obj.spec.weight
    if weight is not None:
        if not isinstance(weight, int) or isinstance(weight, bool):
            raise ValidationError("EndpointGroupBinding: spec.weight must be an integer")
        if not (0 <= weight <= _INT32_MAX):
            raise ValidationError(
                "EndpointGroupBinding: spec.weight must be a non-negative int32"
            )
    if obj.spec.service_ref is not None and not obj.spec.service_ref.name:
        raise ValidationError("EndpointGroupBinding: spec.serviceRef.name is required")
    if obj.spec.ingress_ref is not None and not obj.spec.ingress_ref.name:
        raise ValidationError("EndpointGroupBinding: spec.ingressRef.name is required")
