"""Exceptions are never cached: the failing cached op re-runs on the next
workflow (reference scenario: cached_exception)."""
from lzy_amd import Lzy, op


@op(cache=True, version="1.0")
def raises() -> None:
    print("exception was raised")
    raise ValueError("test")


if __name__ == "__main__":
    try:
        with Lzy().workflow("wf", interactive=False):
            raises()
    except Exception:
        pass
    try:
        with Lzy().workflow("wf", interactive=False):
            raises()
    except Exception:
        pass
