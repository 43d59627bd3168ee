"""Worker-death scenario: rank 1 hard-exits mid-op; the driver must
detect it, fail the affected tasks, and raise instead of hanging."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lzy_amd import Lzy, op
from lzy_amd.exceptions import LzyExecutionError
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime


@op
def maybe_die(i: int) -> int:
    if os.environ.get("RANK") == "1":
        os._exit(7)
    time.sleep(0.2)
    return i


def main() -> None:
    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())
    try:
        with lzy.workflow("death"):
            rs = [maybe_die(i) for i in range(6)]
            [int(r) for r in rs]
        print("DEATH-NOT-DETECTED", flush=True)
    except LzyExecutionError as e:
        print(f"DEATH-DETECTED: {e}", flush=True)
    os._exit(0)


if __name__ == "__main__":
    main()
