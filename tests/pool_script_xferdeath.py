"""Transfer-source death with durable-tier failover (world 3): rank 1
produces a cached value (durable blob written), then hard-exits when
asked to send it to the consumer's rank.  The consumer's settle times
out; the driver recovers the input from the DURABLE TIER onto the
driver rank and re-dispatches — the workflow COMPLETES (reference:
transferFailed → consumer re-pointed at the channel's storage peer,
SlotsService.java:191-240).
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lzy_amd import Lzy, op
from lzy_amd.exceptions import LzyExecutionError
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime

PAYLOAD = 100 * 1024  # >64 KiB (no completion-inline) -> a real transfer


@op(cache=True, version="1.0")
def produce(seed: int) -> bytes:
    # big enough to travel as a transfer, cached -> durable blob exists
    return bytes((seed + i) % 251 for i in range(PAYLOAD))


@op
def consume(data: bytes) -> int:
    return len(data) + data[0] + data[-1]


@op
def pin(r: int) -> int:
    # one trivial op per rank forces round-robin placement so produce
    # and consume land on different non-driver ranks deterministically
    time.sleep(0.05)
    return r


def main() -> None:
    os.environ.setdefault("LZY_SETTLE_WAIT_S", "8")
    if os.environ.get("RANK") == "1":
        from lzy_amd.runtime.pool import WorkerAgent

        orig = WorkerAgent._handle

        def die_on_send(self, msg):
            if msg.get("cmd") == "xfer_send_batch":
                os._exit(9)
            return orig(self, msg)

        WorkerAgent._handle = die_on_send

    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())
    t0 = time.perf_counter()
    try:
        with lzy.workflow("xferdeath"):
            # three produces spread round-robin over the 3 ranks (one is
            # guaranteed on rank 1); six consumers spread likewise, so
            # with near-certainty at least one consumer of rank 1's
            # value sits elsewhere -> a send command reaches rank 1 ->
            # it dies mid-transfer.  Every value is cached (durable
            # blob), so recovery re-sources from the storage tier.
            vs = [produce(s) for s in (3, 4, 5)]
            touch = [int(v2) for v2 in (pin(0), pin(1), pin(2))]  # noqa: F841
            cs = [consume(v) for v in vs for _ in range(2)]
            got = sum(int(c) for c in cs)
        expect = sum(
            2 * (PAYLOAD + (s % 251) + ((s + PAYLOAD - 1) % 251))
            for s in (3, 4, 5)
        )
        assert got == expect, (got, expect)
        elapsed = time.perf_counter() - t0
        print(f"XFERDEATH-RECOVERED in {elapsed:.1f}s", flush=True)
    except LzyExecutionError as e:
        print(f"XFERDEATH-FAILED: {e}", flush=True)
    os._exit(0)


if __name__ == "__main__":
    main()
