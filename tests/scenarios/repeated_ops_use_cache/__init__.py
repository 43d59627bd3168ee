"""Second workflow reuses the cached op result (reference scenario:
repeated_ops_use_cache)."""
from lzy_amd import Lzy, op


@op(cache=True, version="1.0")
def expensive(x: int) -> int:
    print("computing")
    return x * 10


if __name__ == "__main__":
    lzy = Lzy()
    with lzy.workflow("wf", interactive=False):
        print(int(expensive(4)))
    with lzy.workflow("wf", interactive=False):
        print(int(expensive(4)))
