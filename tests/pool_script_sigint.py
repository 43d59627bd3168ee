"""Ctrl-C mid-barrier (world 2): rank 0 receives SIGINT while blocked
in the barrier; the scheduler initiates a StopGraph (queued tasks
cancelled on every rank), drains only the running ops, and the
KeyboardInterrupt propagates — teardown bounded by ~one op duration."""
import os
import signal
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from lzy_amd import Lzy, op
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime

@op
def slow(i: int) -> int:
    time.sleep(1.0)
    return i

@op
def multi(i: int) -> tuple:
    return (i, i * 2)

def main():
    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())
    if os.environ.get("RANK") == "0":
        def fire():
            time.sleep(0.8)
            os.kill(os.getpid(), signal.SIGINT)
        threading.Thread(target=fire, daemon=True).start()
    t0 = time.perf_counter()
    try:
        with lzy.workflow("sigint", interactive=False):
            rs = [slow(i) for i in range(12)]
            vals = [int(r) for r in rs]
        print("SIGINT-NOT-DELIVERED", flush=True)
    except KeyboardInterrupt:
        dt = time.perf_counter() - t0
        assert dt < 6.0, f"drain too slow: {dt:.1f}s"
        print(f"SIGINT-HANDLED in {dt:.1f}s", flush=True)
    os._exit(0)

if __name__ == "__main__":
    main()
