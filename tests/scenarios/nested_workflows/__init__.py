"""A workflow launched from inside an op of another workflow (reference
scenario: nested_workflows — an @op calls run_graph which opens its own
Lzy().workflow())."""
from lzy_amd import Lzy, op


@op
def double(x: int) -> int:
    return x * 2


@op
def run_inner(x: int) -> int:
    with Lzy().workflow("inner", interactive=False):
        return int(double(x))


if __name__ == "__main__":
    with Lzy().workflow("outer", interactive=False):
        a = double(3)
        b = run_inner(int(a))
        print(int(b))
    print("nested ok")
