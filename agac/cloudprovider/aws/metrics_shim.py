"""Tiny indirection so resource managers can count AWS API calls without
importing the metrics module at call sites repeatedly."""

from ...metrics import observe_aws_call


def count(service: str, operation: str):
    observe_aws_call(service, operation)
