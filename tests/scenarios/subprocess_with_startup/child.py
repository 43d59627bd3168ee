import os
import sys

sys.path.insert(0, os.environ.get("LZY_REPO_ROOT", "."))

from lzy_amd import Lzy, op


@op
def triple(x: int) -> int:
    return x * 3


if __name__ == "__main__":
    # launched from inside an op of a (possibly pool) workflow: must run
    # in-process, not try to join the parent's pool rendezvous
    with Lzy().workflow("child-wf", interactive=False):
        print(f"child={int(triple(4))}", flush=True)
