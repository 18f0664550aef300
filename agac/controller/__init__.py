"""The three controllers: GlobalAccelerator, Route53, EndpointGroupBinding
(reference ``pkg/controller/``)."""

from .globalaccelerator import GlobalAcceleratorConfig, GlobalAcceleratorController
from .route53 import Route53Config, Route53Controller
from .endpointgroupbinding import (
    EndpointGroupBindingConfig,
    EndpointGroupBindingController,
)

__all__ = [
    "GlobalAcceleratorConfig",
    "GlobalAcceleratorController",
    "Route53Config",
    "Route53Controller",
    "EndpointGroupBindingConfig",
    "EndpointGroupBindingController",
]
