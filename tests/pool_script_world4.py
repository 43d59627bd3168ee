"""World-size-4 pool scenario on gloo: subgroup gang ops (2 of 4 ranks),
a full-width gang op, fan-out over 4 ranks, chunked cross-rank transfers
(1 MiB chunks), and mixed workflows back-to-back.  Prints WORLD4-OK on
rank 0.  (Run by tests/test_pool_distributed.py.)"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("LZY_CHANNEL_CHUNK_MB", "1")

import torch

from lzy_amd import Lzy, op
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime


@op
def make_tensor(n: int) -> torch.Tensor:
    return torch.arange(n, dtype=torch.float32)


@op
def tsum(t: torch.Tensor) -> float:
    return float(t.sum().item())


@op
def which_rank(i: int) -> int:
    return int(os.environ.get("RANK", "0"))


@op(gpu_count=2)
def gang2(x: float) -> float:
    import torch.distributed as dist

    from lzy_amd.runtime.context import op_context

    ctx = op_context()
    assert ctx.gang_size == 2, ctx
    t = torch.tensor([x])
    dist.all_reduce(t, group=ctx.process_group)
    return float(t.item())


@op(gpu_count=2, cache=True, version="1.0")
def gang_cached(x: float) -> float:
    import torch.distributed as dist

    from lzy_amd.runtime.context import op_context

    ctx = op_context()
    d = os.environ["LZY_AMD_STORAGE"]
    os.makedirs(d, exist_ok=True)
    with open(os.path.join(d, f"gangc.{os.environ.get('RANK')}"), "a") as f:
        f.write("x")
    t = torch.tensor([x + ctx.gang_rank])
    dist.all_reduce(t, group=ctx.process_group)
    return float(t.item())


@op(gpu_count=4)
def gang4(x: float) -> float:
    import torch.distributed as dist

    from lzy_amd.runtime.context import op_context

    ctx = op_context()
    assert ctx.gang_size == 4, ctx
    t = torch.tensor([x * (ctx.gang_rank + 1)])
    dist.all_reduce(t, group=ctx.process_group)
    return float(t.item())


def main() -> None:
    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())

    # fan-out over all 4 ranks
    with lzy.workflow("w4-fanout"):
        ranks = {int(which_rank(i)) for i in range(12)}
        assert ranks == {0, 1, 2, 3}, f"fan-out used {ranks}"

    # multi-chunk cross-rank tensor transfer (3.5 MiB at 1 MiB chunks)
    with lzy.workflow("w4-chunks"):
        n = (3 << 18) + (1 << 17)
        t = make_tensor(n)
        s = tsum(t)
        assert float(s) == float(torch.arange(n, dtype=torch.float32).sum())

    # subgroup gang (2 of 4) then full gang (4 of 4)
    with lzy.workflow("w4-gangs"):
        r2 = gang2(1.5)
        assert float(r2) == 3.0, float(r2)
        r4 = gang4(1.0)
        assert float(r4) == 10.0, float(r4)

    # cached gang op: second workflow serves from the cache (no re-run)
    import glob

    marks = os.environ["LZY_AMD_STORAGE"]
    with lzy.workflow("w4-gc1"):
        a = gang_cached(2.0)
        assert float(a) == 5.0, float(a)  # (2+0) + (2+1)
    runs_before = sum(
        len(open(f).read()) for f in glob.glob(os.path.join(marks, "gangc.*"))
    )
    assert runs_before >= 2, runs_before  # both gang ranks executed once
    with lzy.workflow("w4-gc2"):
        b = gang_cached(2.0)
        assert float(b) == 5.0, float(b)
    runs_after = sum(
        len(open(f).read()) for f in glob.glob(os.path.join(marks, "gangc.*"))
    )
    assert runs_before == runs_after, (runs_before, runs_after)

    if int(os.environ.get("RANK", "0")) == 0:
        print("WORLD4-OK", flush=True)


if __name__ == "__main__":
    main()
