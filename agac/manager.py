"""Controller manager: builds clients and informer factories, starts every
registered controller, then starts the informers
(reference ``pkg/manager/manager.go:22-77`` + the per-controller start funcs
in ``pkg/manager/{globalaccelerator,route53,endpointgroupbinding_controller}.go``).
"""

from __future__ import annotations

import logging
import threading
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from .controller.endpointgroupbinding import (
    EndpointGroupBindingConfig,
    EndpointGroupBindingController,
)
from .controller.globalaccelerator import (
    GlobalAcceleratorConfig,
    GlobalAcceleratorController,
)
from .controller.route53 import Route53Config, Route53Controller
from .kube.informer import SharedInformerFactory

logger = logging.getLogger(__name__)

RESYNC_PERIOD = 30.0  # seconds (reference manager.go:52-53)


@dataclass
class ControllerConfig:
    global_accelerator: GlobalAcceleratorConfig = field(
        default_factory=GlobalAcceleratorConfig
    )
    route53: Route53Config = field(default_factory=Route53Config)
    endpoint_group_binding: EndpointGroupBindingConfig = field(
        default_factory=EndpointGroupBindingConfig
    )


def start_global_accelerator_controller(
    kube_client, informer_factory, config: ControllerConfig, cloud_factory, stop
):
    controller = GlobalAcceleratorController(
        kube_client, informer_factory, config.global_accelerator, cloud_factory
    )
    thread = threading.Thread(
        target=controller.run,
        args=(config.global_accelerator.workers, stop),
        name="global-accelerator-controller",
        daemon=True,
    )
    thread.start()
    return controller, thread


def start_route53_controller(
    kube_client, informer_factory, config: ControllerConfig, cloud_factory, stop
):
    controller = Route53Controller(
        kube_client, informer_factory, config.route53, cloud_factory
    )
    thread = threading.Thread(
        target=controller.run,
        args=(config.route53.workers, stop),
        name="route53-controller",
        daemon=True,
    )
    thread.start()
    return controller, thread


def start_endpoint_group_binding_controller(
    kube_client, informer_factory, config: ControllerConfig, cloud_factory, stop
):
    controller = EndpointGroupBindingController(
        kube_client, informer_factory, config.endpoint_group_binding, cloud_factory
    )
    thread = threading.Thread(
        target=controller.run,
        args=(config.endpoint_group_binding.workers, stop),
        name="endpoint-group-binding-controller",
        daemon=True,
    )
    thread.start()
    return controller, thread


def new_controller_initializers() -> Dict[str, Callable]:
    """Registry of controller start funcs (reference manager.go:34-40)."""
    return {
        "global-accelerator-controller": start_global_accelerator_controller,
        "route53-controller": start_route53_controller,
        "endpoint-group-binding-controller": start_endpoint_group_binding_controller,
    }


class Manager:
    """Owns the informer factory and controller threads for one process."""

    def __init__(self):
        self.controllers: Dict[str, object] = {}
        self.threads: List[threading.Thread] = []
        self.informer_factory: Optional[SharedInformerFactory] = None

    def run(
        self,
        kube_client,
        config: ControllerConfig,
        cloud_factory,
        stop: threading.Event,
        resync_period: float = RESYNC_PERIOD,
        block: bool = True,
    ):
        """Start all controllers and informers.  With ``block=True`` (the
        production path) waits until ``stop`` is set and all controller
        threads exited."""
        informer_factory = SharedInformerFactory(kube_client, resync_period)
        self.informer_factory = informer_factory

        for name, init_fn in new_controller_initializers().items():
            logger.info("Starting %s", name)
            controller, thread = init_fn(
                kube_client, informer_factory, config, cloud_factory, stop
            )
            self.controllers[name] = controller
            self.threads.append(thread)
            logger.info("Started %s", name)

        informer_factory.start(stop)

        if block:
            stop.wait()
            for thread in self.threads:
                thread.join(timeout=5.0)

    def is_ready(self) -> bool:
        """Readiness: every controller's informer caches have synced
        (the /readyz condition for deployment probes)."""
        if self.informer_factory is None:
            return False
        informers = list(self.informer_factory._informers.values())
        return bool(informers) and all(i.has_synced() for i in informers)

    def wait_until_ready(self, timeout: float = 30.0) -> bool:
        """Test helper: wait until every controller's caches have synced and
        workers are pumping."""
        import time

        deadline = time.monotonic() + timeout
        informers = list(self.informer_factory._informers.values())
        while time.monotonic() < deadline:
            if informers and all(i.has_synced() for i in informers):
                return True
            time.sleep(0.01)
        return False
