"""A whole graph of cached ops re-runs with zero executions (reference
scenario: fully_cached_graph)."""
from lzy_amd import Lzy, op


@op(cache=True, version="1.0")
def a(x: int) -> int:
    print("a ran")
    return x + 1


@op(cache=True, version="1.0")
def b(x: int) -> int:
    print("b ran")
    return x * 2


def run() -> None:
    with Lzy().workflow("wf", interactive=False):
        print(int(b(a(1))))


if __name__ == "__main__":
    run()
    run()
    print("finished")
