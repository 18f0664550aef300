"""Fallback metadata for setuptools < 61 (which cannot read pyproject's
[project] table).  Content mirrors pyproject.toml."""

from setuptools import find_packages, setup

setup(
    name="agac",
    version="0.1.0",
    description=(
        "Clean-room Kubernetes controller reconciling Services/Ingresses into "
        "AWS Global Accelerator + Route53, with an EndpointGroupBinding CRD"
    ),
    python_requires=">=3.10",
    packages=find_packages(include=["agac", "agac.*"]),
    install_requires=["click>=8", "pyyaml>=6", "requests>=2.28"],
    extras_require={
        "aws": ["boto3>=1.26"],
        "metrics": ["prometheus_client>=0.16"],
    },
    entry_points={"console_scripts": ["agac = agac.cli:main"]},
)
