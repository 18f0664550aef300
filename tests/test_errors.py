"""Error taxonomy tests (reference pkg/errors/errors_test.go)."""

from agac.errors import NoRetryError, is_no_retry, new_no_retry_errorf


def test_plain_no_retry():
    assert is_no_retry(NoRetryError("boom"))


def test_formatted():
    err = new_no_retry_errorf("invalid resource key: %s", "a/b/c")
    assert is_no_retry(err)
    assert "a/b/c" in str(err)


def test_wrapped_no_retry_detected():
    # errors.As over the wrap chain (reference errors_test.go wrapped case)
    try:
        try:
            raise NoRetryError("inner")
        except NoRetryError as inner:
            raise RuntimeError("outer") from inner
    except RuntimeError as outer:
        assert is_no_retry(outer)


def test_other_errors_retry():
    assert not is_no_retry(RuntimeError("transient"))
    assert not is_no_retry(None)


def test_implicit_context_chain():
    try:
        try:
            raise NoRetryError("inner")
        except NoRetryError:
            raise RuntimeError("outer")  # implicit __context__
    except RuntimeError as outer:
        assert is_no_retry(outer)
