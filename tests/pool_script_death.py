"""Worker-death scenario: rank 1 hard-exits mid-op.

Two modes, selected by LZY_TASK_RETRIES:
  * retries > 0 (default): the driver detects the death, purges the dead
    rank's ownership, and RE-DISPATCHES its inflight tasks onto surviving
    ranks — the workflow COMPLETES (reference failover: storage-peer
    re-pointing + scheduler re-allocation, SlotsService.java:191-240).
  * retries = 0: recovery disabled — the driver must fail the affected
    tasks and raise instead of hanging.
"""
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
    return i * 10


def main() -> None:
    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())
    try:
        with lzy.workflow("death"):
            rs = [maybe_die(i) for i in range(6)]
            vals = [int(r) for r in rs]
        assert vals == [i * 10 for i in range(6)], vals
        # the pool must stay usable AFTER the death: a later workflow
        # schedules only onto surviving ranks
        with lzy.workflow("death-after"):
            vals2 = [int(r) for r in (maybe_die(10), maybe_die(11))]
        assert vals2 == [100, 110], vals2
        print("DEATH-RECOVERED", flush=True)
    except LzyExecutionError as e:
        print(f"DEATH-DETECTED: {e}", flush=True)
    os._exit(0)


if __name__ == "__main__":
    main()
