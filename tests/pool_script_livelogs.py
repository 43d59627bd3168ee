"""Live remote log tail scenario (world 2): an op on rank 1 prints a
marker and keeps running — the marker must appear on the DRIVER console
BEFORE the op finishes (reference: worker→Kafka→client ReadStdSlots
stream while the op runs, KafkaLogsListeners.java:35; client prefixes,
pylzy runtime.py:283-301), not ride the completion event.
"""
import io
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lzy_amd import Lzy, op
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime

OP_RUN_S = 3.0


class _Recorder(io.TextIOBase):
    """Tee recording (timestamp, text) of every driver-console write."""

    def __init__(self, inner):
        self.inner = inner
        self.events = []

    def write(self, s):
        self.events.append((time.perf_counter(), s))
        return self.inner.write(s)

    def flush(self):
        self.inner.flush()


@op
def chatty(i: int) -> tuple:
    print(f"MARKER-{i} up and running")
    time.sleep(OP_RUN_S)
    return os.environ.get("RANK", "?"), time.perf_counter()


def main() -> None:
    rec = _Recorder(sys.stdout)
    if os.environ.get("RANK", "0") == "0":
        sys.stdout = rec  # before the pool installs its capture tee

    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())

    with lzy.workflow("livelogs"):
        rs = [chatty(i) for i in range(2)]
        vals = [tuple(r) for r in rs]

    # at least one op ran on rank 1 (round-robin over 2 ranks)
    remote = [(i, v) for i, v in enumerate(vals) if v[0] == "1"]
    assert remote, f"no op landed on rank 1: {vals}"
    ok = False
    for i, (rank, t_end) in remote:
        seen = [
            t for t, s in rec.events
            if f"MARKER-{i}" in s and "[LZY-chatty]" in s
        ]
        # driver and worker are one machine: perf_counter comparable
        if seen and min(seen) < t_end - 0.5 * OP_RUN_S:
            ok = True
    assert ok, (
        "remote marker did not appear while the op was still running: "
        f"{[(t, s) for t, s in rec.events if 'MARKER' in s]} vs ends "
        f"{remote}"
    )
    sys.stdout = rec.inner
    print("LIVELOGS-OK", flush=True)
    os._exit(0)


if __name__ == "__main__":
    main()
