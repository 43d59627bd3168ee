"""Mid-flight StopGraph scenario: a slow fan-out is aborted from a side
thread while the client thread is blocked in the barrier.  Expectations
(reference: StopGraph RPC workflow-service.proto:12-26 + AbortExecution
semantics, AbortExecution.java:18):

  * queued-but-not-started tasks are cancelled on every rank — teardown
    is bounded by ONE op duration, not the whole queue;
  * the blocked barrier raises WorkflowAbortedError promptly;
  * the pool stays usable — a second workflow completes normally.
"""
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lzy_amd import Lzy, op
from lzy_amd.exceptions import WorkflowAbortedError
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime

OP_SLEEP = 1.2


@op
def slow(i: int) -> int:
    time.sleep(OP_SLEEP)
    return i


@op
def quick(i: int) -> int:
    return i + 100


def main() -> None:
    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())

    aborted_at = {}

    def aborter(wf):
        time.sleep(0.5)
        aborted_at["t"] = time.perf_counter()
        try:
            wf.abort("stop the fan-out")
        except WorkflowAbortedError:
            aborted_at["raised"] = True

    got_abort = False
    th = None
    try:
        with lzy.workflow("stop-fanout") as wf:
            th = threading.Thread(target=aborter, args=(wf,), daemon=True)
            th.start()
            rs = [slow(i) for i in range(16)]
            vals = [int(r) for r in rs]  # blocks in the barrier
    except WorkflowAbortedError:
        got_abort = True
    t_raised = time.perf_counter()
    th.join(timeout=30)

    assert got_abort, "barrier did not raise WorkflowAbortedError"
    assert aborted_at.get("raised"), "side-thread abort did not raise"
    latency = t_raised - aborted_at["t"]
    # 16 ops x 1.2 s on 4 executor slots = 4 waves (~4.8 s) if the queue
    # drains; prompt cancel bounds it to ~one wave + overhead (margin
    # sized for loaded CI boxes)
    assert latency < 2.7 * OP_SLEEP, f"teardown too slow: {latency:.2f}s"
    print(f"STOP-LATENCY {latency:.2f}", flush=True)

    # pool must remain usable after the stop
    with lzy.workflow("after-stop"):
        rs = [quick(i) for i in range(4)]
        vals = [int(r) for r in rs]
    assert vals == [100, 101, 102, 103], vals
    print("STOP-OK", flush=True)
    os._exit(0)


if __name__ == "__main__":
    main()
