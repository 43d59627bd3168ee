"""Worker death DURING a streamed merge plan (world 4): rank 2 hard-
exits exactly when its plan schedule arrives, so peers are already (or
about to be) blocked in chunk recvs.  The driver must fail the plan's
member tasks promptly on the worker_lost signal (survivors' later recv
timeouts surface as stale plan events and are dropped) and the workflow
must raise — not hang."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from lzy_amd import Lzy, op
from lzy_amd.exceptions import LzyExecutionError
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime

N = 1 << 20


@op
def make_shard(i: int) -> torch.Tensor:
    g = torch.Generator().manual_seed(3000 + i)
    return torch.randn(N, generator=g)


@op(pair_reduce=(0.5, 0.5))
def merge(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return (a + b) * 0.5


@op
def total(t: torch.Tensor) -> float:
    return float(t.sum())


def main() -> None:
    if os.environ.get("LZY_STREAM_MERGE", "1") in ("0", "false"):
        # no plans -> nothing to kill mid-plan; scenario vacuous
        print("PLANDEATH-DETECTED (skipped: stream_merge off)", flush=True)
        os._exit(0)
    os.environ.setdefault("LZY_SETTLE_WAIT_S", "15")
    if os.environ.get("RANK") == "2":
        # die exactly when this rank's plan schedule arrives
        from lzy_amd.runtime.pool import WorkerAgent

        WorkerAgent._run_stream_plan = lambda self, msg: os._exit(9)

    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())
    t0 = time.perf_counter()
    try:
        with lzy.workflow("plandeath"):
            shards = [make_shard(i) for i in range(4)]
            m01 = merge(shards[0], shards[1])
            m23 = merge(shards[2], shards[3])
            root = merge(m01, m23)
            float(total(root))
        print("PLANDEATH-NOT-DETECTED", flush=True)
    except LzyExecutionError as e:
        elapsed = time.perf_counter() - t0
        assert elapsed < 60, f"took too long to fail: {elapsed:.1f}s"
        print(f"PLANDEATH-DETECTED after {elapsed:.1f}s: {e}", flush=True)
    os._exit(0)


if __name__ == "__main__":
    main()
