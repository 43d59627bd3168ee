"""Pool-mode crash-resume: run 1 (CRASH=1) hard-exits the whole process
group after stage2's results are persisted; run 2 re-runs the workflow
and resumes from the cache — only stage3 executes.  (Reference:
RestartExecuteGraphTest semantics, on the GpuPoolRuntime.)"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lzy_amd import Lzy, op
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime

MARK_DIR = os.environ["MARK_DIR"]


def mark(name: str) -> None:
    with open(os.path.join(MARK_DIR, f"{name}.r{os.environ.get('RANK','0')}"), "a") as f:
        f.write("x")


@op(cache=True, version="1.0")
def stage1(x: int) -> int:
    mark("stage1")
    return x + 1


@op(cache=True, version="1.0")
def stage2(x: int) -> int:
    mark("stage2")
    return x * 10


@op(cache=True, version="1.0")
def stage3(x: int) -> int:
    mark("stage3")
    return x - 5


def main() -> None:
    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())
    crash = os.environ.get("CRASH") == "1"
    with lzy.workflow("pool-crash-wf") as wf:
        a = stage1(1)
        b = int(stage2(int(a)))  # materializes -> stage1+stage2 persisted
        if crash:
            os._exit(17)
        c = stage3(b)
        print(f"RESULT={int(c)}", flush=True)
    print("POOL-RESUME-OK", flush=True)


if __name__ == "__main__":
    main()
