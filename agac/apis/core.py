"""Core and networking API types: Service, Ingress, Event, Lease.

The subset of k8s.io/api/core/v1 and networking/v1 that the controllers
consume (reference imports in ``pkg/controller/globalaccelerator/controller.go``
and ``pkg/cloudprovider/aws/global_accelerator.go``), in Kubernetes wire
format via ``agac.apis.meta``.
"""

from __future__ import annotations

import typing
from dataclasses import dataclass, field

from .meta import ObjectMeta

SERVICE_TYPE_LOAD_BALANCER = "LoadBalancer"
EVENT_TYPE_NORMAL = "Normal"
EVENT_TYPE_WARNING = "Warning"


# ---------------------------------------------------------------------------
# Service
# ---------------------------------------------------------------------------
@dataclass(slots=True)
class ServicePort:
    name: str = ""
    protocol: str = "TCP"
    port: int = 0
    target_port: typing.Optional[int] = None
    node_port: typing.Optional[int] = None


@dataclass(slots=True)
class ServiceSpec:
    type: str = "ClusterIP"
    ports: typing.List[ServicePort] = field(default_factory=list)
    load_balancer_class: typing.Optional[str] = None
    selector: typing.Dict[str, str] = field(default_factory=dict)


@dataclass(slots=True)
class PortStatus:
    port: int = 0
    protocol: str = "TCP"
    error: typing.Optional[str] = None


@dataclass(slots=True)
class LoadBalancerIngress:
    ip: str = ""
    hostname: str = ""
    ports: typing.List[PortStatus] = field(default_factory=list)


@dataclass(slots=True)
class LoadBalancerStatus:
    ingress: typing.List[LoadBalancerIngress] = field(default_factory=list)


@dataclass(slots=True)
class ServiceStatus:
    load_balancer: LoadBalancerStatus = field(default_factory=LoadBalancerStatus)


@dataclass(slots=True)
class Service:
    kind: typing.ClassVar[str] = "Service"
    api_version: typing.ClassVar[str] = "v1"

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: ServiceSpec = field(default_factory=ServiceSpec)
    status: ServiceStatus = field(default_factory=ServiceStatus)


# ---------------------------------------------------------------------------
# Ingress (networking.k8s.io/v1)
# ---------------------------------------------------------------------------
@dataclass(slots=True)
class ServiceBackendPort:
    name: str = ""
    number: int = 0


@dataclass(slots=True)
class IngressServiceBackend:
    name: str = ""
    port: ServiceBackendPort = field(default_factory=ServiceBackendPort)


@dataclass(slots=True)
class IngressBackend:
    service: typing.Optional[IngressServiceBackend] = None


@dataclass(slots=True)
class HTTPIngressPath:
    path: str = ""
    path_type: str = "Prefix"
    backend: IngressBackend = field(default_factory=IngressBackend)


@dataclass(slots=True)
class HTTPIngressRuleValue:
    paths: typing.List[HTTPIngressPath] = field(default_factory=list)


@dataclass(slots=True)
class IngressRule:
    host: str = ""
    http: typing.Optional[HTTPIngressRuleValue] = None


@dataclass(slots=True)
class IngressSpec:
    ingress_class_name: typing.Optional[str] = None
    default_backend: typing.Optional[IngressBackend] = None
    rules: typing.List[IngressRule] = field(default_factory=list)


@dataclass(slots=True)
class IngressPortStatus:
    port: int = 0
    protocol: str = "TCP"
    error: typing.Optional[str] = None


@dataclass(slots=True)
class IngressLoadBalancerIngress:
    ip: str = ""
    hostname: str = ""
    ports: typing.List[IngressPortStatus] = field(default_factory=list)


@dataclass(slots=True)
class IngressLoadBalancerStatus:
    ingress: typing.List[IngressLoadBalancerIngress] = field(default_factory=list)


@dataclass(slots=True)
class IngressStatus:
    load_balancer: IngressLoadBalancerStatus = field(
        default_factory=IngressLoadBalancerStatus
    )


@dataclass(slots=True)
class Ingress:
    kind: typing.ClassVar[str] = "Ingress"
    api_version: typing.ClassVar[str] = "networking.k8s.io/v1"

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: IngressSpec = field(default_factory=IngressSpec)
    status: IngressStatus = field(default_factory=IngressStatus)


# ---------------------------------------------------------------------------
# Event (recorded by controllers, reference record.EventRecorder)
# ---------------------------------------------------------------------------
@dataclass(slots=True)
class ObjectReference:
    kind: str = ""
    namespace: str = ""
    name: str = ""
    uid: str = ""


@dataclass(slots=True)
class EventSource:
    component: str = ""


@dataclass(slots=True)
class Event:
    kind: typing.ClassVar[str] = "Event"
    api_version: typing.ClassVar[str] = "v1"

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    involved_object: ObjectReference = field(default_factory=ObjectReference)
    reason: str = ""
    message: str = ""
    type: str = EVENT_TYPE_NORMAL
    source: EventSource = field(default_factory=EventSource)
    count: int = 1
    first_timestamp: typing.Optional[str] = None
    last_timestamp: typing.Optional[str] = None


# ---------------------------------------------------------------------------
# Lease (coordination.k8s.io/v1, used by leader election)
# ---------------------------------------------------------------------------
@dataclass(slots=True)
class LeaseSpec:
    holder_identity: typing.Optional[str] = None
    lease_duration_seconds: typing.Optional[int] = None
    acquire_time: typing.Optional[str] = None
    renew_time: typing.Optional[str] = None
    lease_transitions: int = 0


@dataclass(slots=True)
class Lease:
    kind: typing.ClassVar[str] = "Lease"
    api_version: typing.ClassVar[str] = "coordination.k8s.io/v1"

    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: LeaseSpec = field(default_factory=LeaseSpec)
