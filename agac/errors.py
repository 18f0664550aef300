"""Error taxonomy for the reconcile engine.

Mirrors the reference's ``pkg/errors/errors.go:8-39``: a ``NoRetryError``
marks reconcile failures that must NOT be requeued (the engine Forget()s the
key instead of AddRateLimited).  ``is_no_retry`` walks the ``__cause__``
chain, matching Go's ``errors.As`` over wrapped errors.
"""

from __future__ import annotations


class NoRetryError(Exception):
    """An error for which the reconcile engine must not retry the key."""


def new_no_retry_errorf(fmt: str, *args) -> NoRetryError:
    """Convenience formatter, reference ``pkg/errors/errors.go:19-23``."""
    return NoRetryError(fmt % args if args else fmt)


def is_no_retry(err: BaseException | None) -> bool:
    """True if ``err`` or anything in its cause/context chain is NoRetryError.

    Reference ``pkg/errors/errors.go:34-39`` (``errors.As`` over Unwrap).
    """
    seen = set()
    while err is not None and id(err) not in seen:
        if isinstance(err, NoRetryError):
            return True
        seen.add(id(err))
        err = err.__cause__ if err.__cause__ is not None else err.__context__
    return False
