"""The flagship bench's 8-stage DAG at world_size 4 on CPU/gloo — the
same shape the driver scales to 8 GPUs (bench.py run_dag), validated
multi-rank without a GPU: fan-out x4, cross-rank tree merge, scalar
gather.  Prints BENCH-DAG-OK on rank 0."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("LZY_BENCH_SHARD_MB", "2")

import bench  # noqa: E402  (repo-root bench.py)
from lzy_amd import Lzy  # noqa: E402
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime  # noqa: E402


def main() -> None:
    pool = GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())
    width = int(os.environ.get("WORLD_SIZE", "4"))
    for step in range(3):
        out = bench.run_dag(lzy, width, step)
        assert isinstance(out, float), out
    pool.sync_all()
    print("BENCH-DAG-OK", flush=True)


if __name__ == "__main__":
    main()
