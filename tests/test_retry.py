"""Retry utility tests (reference: util-db withRetries semantics)."""
import pytest

from lzy_amd.utils.retry import retry, with_retries


def test_succeeds_after_transient_failures():
    calls = []

    def flaky():
        calls.append(1)
        if len(calls) < 3:
            raise OSError("transient")
        return "ok"

    assert with_retries(flaky, base_delay_s=0.001) == "ok"
    assert len(calls) == 3


def test_exhausts_and_reraises():
    def always():
        raise OSError("permanent")

    with pytest.raises(OSError, match="permanent"):
        with_retries(always, attempts=3, base_delay_s=0.001)


def test_non_retryable_propagates_immediately():
    calls = []

    def boom():
        calls.append(1)
        raise ValueError("logic error")

    with pytest.raises(ValueError):
        with_retries(boom, base_delay_s=0.001)
    assert len(calls) == 1


def test_decorator():
    calls = []

    @retry(attempts=4, base_delay_s=0.001)
    def f(x):
        calls.append(x)
        if len(calls) < 2:
            raise OSError("once")
        return x * 2

    assert f(5) == 10
    assert calls == [5, 5]
