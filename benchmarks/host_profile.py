"""cProfile of run_dag host-side on the GPU (find the non-device gap)."""
import cProfile
import io
import os
import pstats
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("LZY_AMD_STORAGE", "/tmp/lzy_hostprof")

import bench
from lzy_amd import Lzy
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime


def main() -> None:
    pool = GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())
    for i in range(4):
        bench.run_dag(lzy, 1, i)
    pr = cProfile.Profile()
    pr.enable()
    for i in range(20):
        bench.run_dag(lzy, 1, 100 + i)
    pr.disable()
    s = io.StringIO()
    pstats.Stats(pr, stream=s).sort_stats("cumulative").print_stats(35)
    print(s.getvalue())


if __name__ == "__main__":
    main()
