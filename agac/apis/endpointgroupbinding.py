"""EndpointGroupBinding CRD types, group ``operator.h3poteto.dev/v1alpha1``.

Reference ``pkg/apis/endpointgroupbinding/v1alpha1/types.go:16-70``.  The
JSON field names (incl. ``clientIPPreservation``) are part of the CRD wire
format and must match the reference exactly.
"""

from __future__ import annotations

import typing
from dataclasses import dataclass, field

from .meta import ObjectMeta

GROUP = "operator.h3poteto.dev"
VERSION = "v1alpha1"
FINALIZER = "operator.h3poteto.dev/endpointgroupbindings"


@dataclass(slots=True)
class ServiceReference:
    name: str = ""


@dataclass(slots=True)
class IngressReference:
    name: str = ""


@dataclass(slots=True)
class EndpointGroupBindingSpec:
    _json_overrides: typing.ClassVar[dict] = {
        "client_ip_preservation": "clientIPPreservation",
    }

    endpoint_group_arn: str = ""
    client_ip_preservation: bool = False
    weight: typing.Optional[int] = None
    service_ref: typing.Optional[ServiceReference] = None
    ingress_ref: typing.Optional[IngressReference] = None


@dataclass(slots=True)
class EndpointGroupBindingStatus:
    _keep_empty: typing.ClassVar[set] = {"endpoint_ids"}

    endpoint_ids: typing.List[str] = field(default_factory=list)
    observed_generation: int = 0


@dataclass(slots=True)
class EndpointGroupBinding:
    kind: typing.ClassVar[str] = "EndpointGroupBinding"
    api_version: typing.ClassVar[str] = f"{GROUP}/{VERSION}"

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: EndpointGroupBindingSpec = field(default_factory=EndpointGroupBindingSpec)
    status: EndpointGroupBindingStatus = field(
        default_factory=EndpointGroupBindingStatus
    )
