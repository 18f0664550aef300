"""Canonical test fixtures (reference ``pkg/fixture/endpointgroupbinding.go``)."""

from __future__ import annotations

from typing import Optional

from .apis import endpointgroupbinding as egb
from .apis.meta import ObjectMeta


def endpoint_group_binding(
    name: str = "test",
    namespace: str = "default",
    endpoint_group_arn: str = (
        "arn:aws:globalaccelerator::123456789012:accelerator/"
        "11111111-2222-3333-4444-555555555555/listener/aaaaaaaa/"
        "endpoint-group/bbbbbbbb"
    ),
    weight: Optional[int] = 128,
    client_ip_preservation: bool = False,
    service_name: Optional[str] = "test-service",
    ingress_name: Optional[str] = None,
) -> egb.EndpointGroupBinding:
    """The canonical EndpointGroupBinding used by unit + e2e tests."""
    spec = egb.EndpointGroupBindingSpec(
        endpoint_group_arn=endpoint_group_arn,
        client_ip_preservation=client_ip_preservation,
        weight=weight,
    )
    if service_name is not None:
        spec.service_ref = egb.ServiceReference(name=service_name)
    if ingress_name is not None:
        spec.ingress_ref = egb.IngressReference(name=ingress_name)
        spec.service_ref = None
    return egb.EndpointGroupBinding(
        metadata=ObjectMeta(name=name, namespace=namespace), spec=spec
    )
