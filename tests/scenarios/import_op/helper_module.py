"""Local module imported by the scenario — its ops and helpers must reach
workers (reference scenario: import — local-modules sync)."""
from lzy_amd import op

FACTOR = 7


def scale(x: int) -> int:
    return x * FACTOR


@op
def imported_op(x: int) -> int:
    return scale(x) + 1
