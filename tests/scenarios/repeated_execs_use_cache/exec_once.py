import os
import sys

sys.path.insert(0, os.environ.get("LZY_REPO_ROOT", "."))

from lzy_amd import Lzy, op


@op(cache=True, version="1.0")
def expensive(x: int) -> int:
    print("computing", flush=True)
    return x * 11


if __name__ == "__main__":
    with Lzy().workflow("wf", interactive=False):
        print(int(expensive(3)), flush=True)
