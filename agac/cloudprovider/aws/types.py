"""AWS resource shapes used by the resource managers.

The subset of aws-sdk-go-v2 types the reference touches
(``gatypes.Accelerator``/``Listener``/``EndpointGroup``,
``elbv2types.LoadBalancer``, ``route53types.HostedZone``/``ResourceRecordSet``).
"""

from __future__ import annotations

import typing
from dataclasses import dataclass, field

# Enum-ish constants (string unions in the SDK)
PROTOCOL_TCP = "TCP"
PROTOCOL_UDP = "UDP"
CLIENT_AFFINITY_NONE = "NONE"
IP_ADDRESS_TYPE_IPV4 = "IPV4"
IP_ADDRESS_TYPE_DUAL_STACK = "DUAL_STACK"
ACCELERATOR_STATUS_DEPLOYED = "DEPLOYED"
ACCELERATOR_STATUS_IN_PROGRESS = "IN_PROGRESS"
LB_STATE_ACTIVE = "active"
LB_STATE_PROVISIONING = "provisioning"
LB_STATE_FAILED = "failed"
RR_TYPE_A = "A"
RR_TYPE_TXT = "TXT"
CHANGE_ACTION_CREATE = "CREATE"
CHANGE_ACTION_UPSERT = "UPSERT"
CHANGE_ACTION_DELETE = "DELETE"

# Alias hosted zone of every Global Accelerator (fixed by AWS; reference
# route53.go:255 hardcodes it the same way).
GLOBAL_ACCELERATOR_HOSTED_ZONE_ID = "Z2BJ6XQ5FK7U4H"


@dataclass(slots=True)
class Tag:
    key: str
    value: str


@dataclass(slots=True)
class Accelerator:
    accelerator_arn: str = ""
    name: str = ""
    dns_name: str = ""
    enabled: bool = True
    status: str = ACCELERATOR_STATUS_DEPLOYED
    ip_address_type: str = IP_ADDRESS_TYPE_DUAL_STACK


@dataclass(slots=True)
class PortRange:
    from_port: int = 0
    to_port: int = 0


@dataclass(slots=True)
class Listener:
    listener_arn: str = ""
    port_ranges: typing.List[PortRange] = field(default_factory=list)
    protocol: str = PROTOCOL_TCP
    client_affinity: str = CLIENT_AFFINITY_NONE


@dataclass(slots=True)
class EndpointDescription:
    endpoint_id: str = ""
    weight: typing.Optional[int] = None
    client_ip_preservation_enabled: typing.Optional[bool] = None
    health_state: str = "HEALTHY"


@dataclass(slots=True)
class EndpointGroup:
    endpoint_group_arn: str = ""
    endpoint_group_region: str = ""
    endpoint_descriptions: typing.List[EndpointDescription] = field(default_factory=list)


@dataclass(slots=True)
class EndpointConfiguration:
    endpoint_id: str = ""
    weight: typing.Optional[int] = None
    client_ip_preservation_enabled: typing.Optional[bool] = None


@dataclass(slots=True)
class LoadBalancer:
    load_balancer_arn: str = ""
    load_balancer_name: str = ""
    dns_name: str = ""
    state_code: str = LB_STATE_ACTIVE
    type: str = "network"  # "network" (NLB) or "application" (ALB)
    scheme: str = "internet-facing"


@dataclass(slots=True)
class HostedZone:
    id: str = ""
    name: str = ""  # always dot-terminated, e.g. "example.com."


@dataclass(slots=True)
class AliasTarget:
    dns_name: str = ""
    evaluate_target_health: bool = True
    hosted_zone_id: str = ""


@dataclass(slots=True)
class ResourceRecord:
    value: str = ""


@dataclass(slots=True)
class ResourceRecordSet:
    name: str = ""  # dot-terminated, wildcards octal-escaped ("\\052")
    type: str = RR_TYPE_A
    ttl: typing.Optional[int] = None
    resource_records: typing.List[ResourceRecord] = field(default_factory=list)
    alias_target: typing.Optional[AliasTarget] = None


@dataclass(slots=True)
class Change:
    action: str
    record_set: ResourceRecordSet
