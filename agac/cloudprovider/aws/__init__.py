"""AWS cloud provider: resource managers for ELBv2, Global Accelerator and
Route53 (reference ``pkg/cloudprovider/aws``)."""

from .types import (
    Accelerator,
    AliasTarget,
    EndpointDescription,
    EndpointGroup,
    HostedZone,
    Listener,
    LoadBalancer,
    PortRange,
    ResourceRecord,
    ResourceRecordSet,
    Tag,
)
from .errors import (
    AWSAPIError,
    EndpointGroupNotFoundException,
    ListenerNotFoundException,
    error_code,
    is_error_code,
)
from .load_balancer import get_lb_name_from_hostname, get_region_from_arn
from .client import AWS, CloudFactory

__all__ = [
    "AWS",
    "CloudFactory",
    "Accelerator",
    "AliasTarget",
    "AWSAPIError",
    "EndpointDescription",
    "EndpointGroup",
    "EndpointGroupNotFoundException",
    "HostedZone",
    "Listener",
    "ListenerNotFoundException",
    "LoadBalancer",
    "PortRange",
    "ResourceRecord",
    "ResourceRecordSet",
    "Tag",
    "error_code",
    "is_error_code",
    "get_lb_name_from_hostname",
    "get_region_from_arn",
]
