"""Typed AWS API errors.

Mirror of smithy-go APIError + the typed not-found exceptions the reference
branches on (``gatypes.ListenerNotFoundException`` in
``global_accelerator.go:300``, ``EndpointGroupNotFoundException`` in
``egb/reconcile.go:56`` via error-code string match).
"""

from __future__ import annotations


class AWSAPIError(Exception):
    """An AWS API error with a smithy-style error code."""

    code = "InternalError"

    def __init__(self, message: str = "", code: str | None = None):
        super().__init__(message or self.code)
        if code is not None:
            self.code = code


class AcceleratorNotFoundException(AWSAPIError):
    code = "AcceleratorNotFoundException"


class ListenerNotFoundException(AWSAPIError):
    code = "ListenerNotFoundException"


class EndpointGroupNotFoundException(AWSAPIError):
    code = "EndpointGroupNotFoundException"


class LoadBalancerNotFoundException(AWSAPIError):
    code = "LoadBalancerNotFound"


class AcceleratorNotDisabledException(AWSAPIError):
    """DeleteAccelerator on an enabled accelerator (real AWS behavior; this
    is why the reference disables + polls before deleting,
    ``global_accelerator.go:743-784``)."""

    code = "AcceleratorNotDisabledException"


class NoSuchHostedZone(AWSAPIError):
    code = "NoSuchHostedZone"


class InvalidChangeBatch(AWSAPIError):
    code = "InvalidChangeBatch"


ERR_ENDPOINT_GROUP_NOT_FOUND = "EndpointGroupNotFoundException"


def error_code(err: BaseException) -> str | None:
    """smithy.APIError.ErrorCode() equivalent."""
    if isinstance(err, AWSAPIError):
        return err.code
    return None


def is_error_code(err: BaseException, code: str) -> bool:
    return error_code(err) == code
