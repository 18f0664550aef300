"""The AWS client bundle and cloud factory.

Reference ``pkg/cloudprovider/aws/aws.go:12-38``: ``AWS{lb, ga, route53}``
constructed per reconcile via ``NewAWS(region)``; the Global Accelerator and
Route53 clients are always homed in us-west-2 (GA is a global service),
while the ELBv2 client is regional.

The controllers take a ``CloudFactory`` (``factory(region) -> AWS``) instead
of hard-constructing the SDK — this is the mockable seam BASELINE.json's
test matrix requires.  ``FakeCloudFactory`` binds every region to one shared
``FakeAWSBackend``; a boto3-backed factory can be slotted in for production
without touching any controller code.
"""

from __future__ import annotations

import time
from typing import Callable, Optional

from .global_accelerator import GlobalAcceleratorMixin
from .load_balancer import LoadBalancerMixin
from .route53 import Route53Mixin


class AWS(LoadBalancerMixin, GlobalAcceleratorMixin, Route53Mixin):
    """Resource managers over the three service clients.

    ``lb``/``ga``/``route53`` expose the AWS operation surface (implemented
    by ``agac.cloudprovider.fake`` in tests, or a real-SDK adapter in
    production).  ``poll_interval``/``poll_timeout`` control the
    disable→delete status poll (reference hardcodes 10s/3min,
    ``global_accelerator.go:756``); ``sleep`` is injectable so tests run the
    poll loop without wall-clock delays.
    """

    def __init__(
        self,
        lb,
        ga,
        route53,
        region: str,
        poll_interval: float = 10.0,
        poll_timeout: float = 180.0,
        sleep: Callable[[float], None] = time.sleep,
        lb_not_active_retry: float = 30.0,
        ga_missing_retry: float = 60.0,
    ):
        self.lb = lb
        self.ga = ga
        self.route53 = route53
        self.region = region
        self.poll_interval = poll_interval
        self.poll_timeout = poll_timeout
        self.sleep = sleep
        # requeue intervals (reference hardcodes 30s / 60s; injectable here
        # so hermetic tests exercise the retry paths without wall-clock waits)
        self.lb_not_active_retry = lb_not_active_retry
        self.ga_missing_retry = ga_missing_retry


# factory(region) -> AWS
CloudFactory = Callable[[str], AWS]


class FakeCloudFactory:
    """Binds every region to a single shared in-memory backend."""

    def __init__(
        self,
        backend=None,
        poll_interval: float = 0.0,
        poll_timeout: float = 5.0,
        sleep: Optional[Callable[[float], None]] = None,
        lb_not_active_retry: float = 30.0,
        ga_missing_retry: float = 60.0,
    ):
        if backend is None:
            from ..fake import FakeAWSBackend

            backend = FakeAWSBackend()
        self.backend = backend
        self.poll_interval = poll_interval
        self.poll_timeout = poll_timeout
        self.sleep = sleep if sleep is not None else (lambda s: None)
        self.lb_not_active_retry = lb_not_active_retry
        self.ga_missing_retry = ga_missing_retry

    def __call__(self, region: str) -> AWS:
        from ..fake.backend import RegionalELBv2View

        return AWS(
            lb=RegionalELBv2View(self.backend.elbv2, region),
            ga=self.backend.ga,
            route53=self.backend.route53,
            region=region,
            poll_interval=self.poll_interval,
            poll_timeout=self.poll_timeout,
            sleep=self.sleep,
            lb_not_active_retry=self.lb_not_active_retry,
            ga_missing_retry=self.ga_missing_retry,
        )


def boto3_cloud_factory() -> CloudFactory:
    """Production factory backed by boto3 (not installed in this image);
    kept as the explicit production seam."""
    try:
        import boto3  # noqa: F401
    except ImportError as e:  # pragma: no cover
        raise RuntimeError(
            "boto3 is required for the real AWS cloud provider; "
            "install boto3 or inject a CloudFactory"
        ) from e
    from .boto3_adapter import new_boto3_factory  # pragma: no cover

    return new_boto3_factory()  # pragma: no cover
