"""Retry with exponential backoff (reference: util-db ``withRetries``
DbHelper.java:25-33 and util-grpc retry configs GrpcUtils.java:31-79 —
transient-failure retries around storage/IO operations)."""
from __future__ import annotations

import logging
import random
import time
from functools import wraps
from typing import Callable, Tuple, Type, TypeVar

_LOG = logging.getLogger("lzy_amd.retry")

T = TypeVar("T")


def with_retries(
    fn: Callable[[], T],
    *,
    attempts: int = 5,
    base_delay_s: float = 0.05,
    max_delay_s: float = 2.0,
    retry_on: Tuple[Type[BaseException], ...] = (OSError,),
    what: str = "",
) -> T:
    """Run ``fn`` with up to ``attempts`` tries; exponential backoff with
    full jitter between tries.  Re-raises the last error."""
    delay = base_delay_s
    for attempt in range(1, attempts + 1):
        try:
            return fn()
        except retry_on as e:
            if attempt == attempts:
                raise
            sleep = random.uniform(0, min(delay, max_delay_s))
            _LOG.warning(
                "retry %d/%d for %s after %s: %s",
                attempt, attempts, what or getattr(fn, "__name__", "op"),
                type(e).__name__, e,
            )
            time.sleep(sleep)
            delay *= 2
    raise AssertionError("unreachable")


def retry(
    *,
    attempts: int = 5,
    base_delay_s: float = 0.05,
    max_delay_s: float = 2.0,
    retry_on: Tuple[Type[BaseException], ...] = (OSError,),
):
    """Decorator form of :func:`with_retries`."""

    def deco(f: Callable[..., T]) -> Callable[..., T]:
        @wraps(f)
        def wrapper(*args, **kwargs) -> T:
            return with_retries(
                lambda: f(*args, **kwargs),
                attempts=attempts,
                base_delay_s=base_delay_s,
                max_delay_s=max_delay_s,
                retry_on=retry_on,
                what=f.__qualname__,
            )

        return wrapper

    return deco
