"""Diamond + fan-out graph over mixed types (reference scenario:
pylzy/tests/scenarios/complex_graph)."""
from typing import List

from lzy_amd import Lzy, op


@op
def source(n: int) -> List[int]:
    return list(range(n))


@op
def square_all(xs: List[int]) -> List[int]:
    return [x * x for x in xs]


@op
def total(xs: List[int]) -> int:
    return sum(xs)


@op
def describe(sq_sum: int, raw_sum: int) -> str:
    return f"squares={sq_sum} raw={raw_sum}"


@op
def shout(s: str) -> str:
    return s.upper()


if __name__ == "__main__":
    lzy = Lzy()
    with lzy.workflow("complex_graph", interactive=False):
        xs = source(5)
        sq = square_all(xs)
        msg = describe(total(sq), total(xs))
        loud = shout(msg)
        print(msg)
        print(loud)
    print("done")
