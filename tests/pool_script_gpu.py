"""GPU pool scenario for 2 ranks (possibly sharing one GPU in test)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from lzy_amd import Lzy, op
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime


@op
def make_dev(n: int, mark: int) -> torch.Tensor:
    from lzy_amd.ops import fill_pattern

    t = torch.empty(n, device="cuda", dtype=torch.float32)
    fill_pattern(t, seed=mark)
    t.view(torch.int32).bitwise_and_(0x3FFFFFFF)
    return t


@op
def rank_of() -> int:
    return int(os.environ.get("RANK", "0"))


@op
def combine(a: torch.Tensor, b: torch.Tensor) -> float:
    # consumes tensors possibly produced on the other rank
    assert a.is_cuda and b.is_cuda
    return float((a.float() - b.float()).abs().sum().item())


@op
def hash_dev(t: torch.Tensor) -> int:
    from lzy_amd.ops import device_checksum

    return device_checksum(t)


def main() -> None:
    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())

    with lzy.workflow("gpu-pool-1") as wf:
        # force both ranks to produce (fan-out placement)
        rs = [rank_of() for _ in range(6)]
        assert {int(r) for r in rs} == {0, 1}, f"ranks {[int(r) for r in rs]}"

    with lzy.workflow("gpu-pool-2") as wf:
        a = make_dev(1 << 20, 1)
        b = make_dev(1 << 20, 2)
        d = combine(a, b)  # at least one input crosses ranks sometimes
        h1 = hash_dev(a)
        h2 = hash_dev(a)
        assert int(h1) == int(h2)
        assert float(d) >= 0.0

    print("POOL-GPU-OK", flush=True)


if __name__ == "__main__":
    main()
