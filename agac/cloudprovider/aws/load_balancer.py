"""ELB hostname parsing and ARN helpers.

Behavior parity with reference ``pkg/cloudprovider/aws/load_balancer.go``:

- ALB hostnames look like ``name-hash.region.elb.amazonaws.com`` (with an
  optional ``internal-`` prefix on the first label);
- NLB hostnames look like ``name-hash.elb.region.amazonaws.com``;
- the dispatch order matters: the ALB pattern is ``\\.elb\\.amazonaws\\.com$``
  (i.e. the label right before amazonaws.com is "elb") and is tried first,
  then the NLB pattern ``\\.elb\\..+\\.amazonaws\\.com$``
  (reference ``load_balancer.go:32-44``).
"""

from __future__ import annotations

import re
from typing import Tuple

_ALB_SUFFIX = re.compile(r"\.elb\.amazonaws\.com$")
_NLB_SUFFIX = re.compile(r"\.elb\..+\.amazonaws\.com$")
_INTERNAL_PREFIX = re.compile(r"^internal-")
_INTERNAL_ALB_NAME = re.compile(r"^internal\-([\w\-]+)\-[\w]+$")
_LB_NAME = re.compile(r"^([\w\-]+)\-[\w]+$")


def get_lb_name_from_hostname(hostname: str) -> Tuple[str, str]:
    """Returns (lb_name, region) parsed from an ELB hostname; raises
    ValueError for non-ELB hostnames (reference ``GetLBNameFromHostname``)."""
    if _ALB_SUFFIX.search(hostname):
        return _match_alb_hostname(hostname)
    if _NLB_SUFFIX.search(hostname):
        return _match_nlb_hostname(hostname)
    raise ValueError(f"{hostname} is not Elastic Load Balancer")


def _match_alb_hostname(hostname: str) -> Tuple[str, str]:
    labels = hostname.split(".")
    subdomain, region = labels[0], labels[1]
    if _INTERNAL_PREFIX.search(subdomain):
        m = _INTERNAL_ALB_NAME.findall(subdomain)
        if len(m) != 1:
            raise ValueError(f"Failed to parse subdomain for internal ALB: {subdomain}")
        return m[0], region
    m = _LB_NAME.findall(subdomain)
    if len(m) != 1:
        raise ValueError(f"Failed to parse subdomain for public ALB: {subdomain}")
    return m[0], region


def _match_nlb_hostname(hostname: str) -> Tuple[str, str]:
    labels = hostname.split(".")
    subdomain, region = labels[0], labels[2]
    m = _LB_NAME.findall(subdomain)
    if len(m) != 1:
        raise ValueError(f"Failed to parse subdomain for NLB: {subdomain}")
    return m[0], region


def get_region_from_arn(arn: str) -> str:
    """4th colon-separated field of an ARN (reference ``GetRegionFromARN``)."""
    return arn.split(":")[3]


class LoadBalancerMixin:
    """ELBv2 resource manager methods (reference load_balancer.go:13-30)."""

    def get_load_balancer(self, name: str):
        """DescribeLoadBalancers(Names=[name]) → the matching LB.

        Raises LoadBalancerNotFound (typed, from the API) or ValueError if
        the response somehow lacks the requested name."""
        from ...metrics import observe_aws_call

        observe_aws_call("elbv2", "DescribeLoadBalancers")
        lbs, _ = self.lb.describe_load_balancers(names=[name])
        for lb in lbs:
            if lb.load_balancer_name == name:
                return lb
        raise ValueError(f"Could not find LoadBalancer: {name}")
