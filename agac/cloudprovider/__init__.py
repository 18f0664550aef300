"""Cloud-provider seam.

``detect_cloud_provider`` mirrors reference ``pkg/cloudprovider/provider.go:8-17``.
The AWS resource managers live in ``agac.cloudprovider.aws``; the stateful
in-memory AWS used by tests/bench lives in ``agac.cloudprovider.fake``.
"""

from __future__ import annotations


def detect_cloud_provider(hostname: str) -> str:
    """Returns the provider name for an LB hostname, or raises ValueError.

    The last two DNS labels decide: ``*.amazonaws.com`` ⇒ "aws"
    (reference ``provider.go:8-17``).
    """
    parts = hostname.split(".")
    if len(parts) >= 2:
        domain = parts[-2] + "." + parts[-1]
        if domain == "amazonaws.com":
            return "aws"
    else:
        domain = hostname
    raise ValueError(f"Unknown cloud provider: {domain}")
