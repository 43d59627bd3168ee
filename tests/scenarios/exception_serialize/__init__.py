"""A custom exception type raised in an op crosses the process/data-plane
boundary intact (reference scenario: exception_serialize)."""
from lzy_amd import Lzy, op
from lzy_amd.exceptions import LzyExecutionError


class DomainError(Exception):
    def __init__(self, code: int, msg: str):
        super().__init__(msg)
        self.code = code


@op
def fails(x: int) -> int:
    raise DomainError(42, f"bad input {x}")


if __name__ == "__main__":
    try:
        with Lzy().workflow("wf", interactive=False):
            fails(5)
    except LzyExecutionError as e:
        c = e.__cause__
        print(type(c).__name__)
        print(c.code)
        print(c)
