"""Framework-floor probe: the flagship DAG's SHAPE with no-op bodies.

Measures what the runtime itself costs per DAG (dispatch, chaining,
settle, store, events, journal, barriers) with zero device work — the
lower bound the real bench's host-side gap is judged against.

Usage: python benchmarks/floor_probe.py [dags]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("LZY_AMD_STORAGE", "/tmp/lzy_floorprobe")

from lzy_amd import Lzy, op
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime


@op
def nop0(i: int) -> int:
    return i


@op
def nop1(x: int) -> int:
    return x


@op
def nop2(x: int) -> int:
    return x


@op
def fan1(x: int) -> int:
    return x


@op
def fan2(x: int) -> int:
    return x


@op
def tail1(x: int) -> int:
    return x


@op
def gather(a: int, b: int, c: int) -> int:
    return a + b + c


def run_dag(lzy, i):
    # same shape as the flagship at width 1: chain of 3, fan-out of 2,
    # chain tail, 3-way gather — 7 ops, 2 barrier batches
    with lzy.workflow(f"floor-{i}", interactive=False):
        a = nop2(nop1(nop0(i)))
        f1, f2 = fan1(a), fan2(a)
        t = tail1(f1)
        s = int(gather(t, f2, a))
    return s


def main() -> None:
    dags = int(sys.argv[1]) if len(sys.argv) > 1 else 200
    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())
    for i in range(10):
        run_dag(lzy, i)
    t0 = time.perf_counter()
    for i in range(dags):
        assert run_dag(lzy, 100 + i) == 3 * (100 + i)
    dt = (time.perf_counter() - t0) / dags
    from lzy_amd.utils.metrics import METRICS

    d = METRICS.timing_stats("lzy_dispatch")
    o = METRICS.timing_stats("lzy_task_overhead")
    print(
        f"floor: {dt*1e3:.3f} ms/DAG (7 no-op tasks, 2 batches); "
        f"dispatch mean {d.get('mean', 0)*1e6:.0f} us, "
        f"task roundtrip-minus-op mean {o.get('mean', 0)*1e3:.3f} ms"
    )


if __name__ == "__main__":
    main()
