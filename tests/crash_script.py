"""Crash-resume scenario (reference: LzyServiceRestartTests /
RestartExecuteGraphTest — kill the service mid-graph, restart, assert the
durable machinery resumes without re-doing completed work).

Run 1 (CRASH=1): the process dies (os._exit) after the second op
completes.  Run 2: the workflow re-runs; ops 1-2 are served from the
result cache (their outputs were persisted before the crash), only op 3
executes.  Side-effect markers in MARK_DIR record actual executions.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lzy_amd import Lzy, op
from lzy_amd.runtime.local import LocalRuntime
from lzy_amd.utils.faults import FAULTS

MARK_DIR = os.environ["MARK_DIR"]


def mark(name: str) -> None:
    with open(os.path.join(MARK_DIR, name), "a") as f:
        f.write("x")


@op(cache=True, version="1.0")
def stage1(x: int) -> int:
    mark("stage1")
    return x + 1


@op(cache=True, version="1.0")
def stage2(x: int) -> int:
    mark("stage2")
    return x * 2


@op(cache=True, version="1.0")
def stage3(x: int) -> int:
    mark("stage3")
    return x - 3


def main() -> None:
    at = os.environ.get("CRASH_AT")
    if at is not None:
        # die after the (at+1)-th EXECUTED op — cached ops do not count,
        # so repeated runs with CRASH_AT=0 walk the crash point forward
        # one stage per restart (reference stress-runner restart chain)
        FAULTS.arm("executor.after_run", countdown=int(at), kind="exit")
    elif os.environ.get("CRASH") == "1":
        # kill the process right after the second op persists its result
        FAULTS.arm("executor.after_run", countdown=1, kind="exit")

    lzy = Lzy(runtime=LocalRuntime())
    with lzy.workflow("crashy") as wf:
        a = stage1(10)
        b = stage2(a)
        c = stage3(b)
        assert int(c) == 19
    print("CRASH-SCRIPT-DONE", flush=True)


if __name__ == "__main__":
    main()
