"""Distributed pool scenario — run under torch.distributed.run with
world_size 2 on CPU (gloo).  Exercises: cross-rank placement & transfers
(tensor + pickled objects), gang ops with in-op collectives, failure
transport, whiteboards, multiple sequential workflows.

Executed by tests/test_pool_distributed.py; prints POOL-SCRIPT-OK on
success (rank 0).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from lzy_amd import Lzy, op, whiteboard_
from lzy_amd.exceptions import LzyExecutionError
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime
from dataclasses import dataclass


@op
def inc(x: int) -> int:
    return x + 1


@op
def make_tensor(n: int) -> torch.Tensor:
    return torch.arange(n, dtype=torch.float32)


@op
def tensor_sum(t: torch.Tensor) -> float:
    return float(t.sum().item())


@op
def which_rank(x: int) -> int:
    return int(os.environ.get("RANK", "0"))


@op(gpu_count=2)
def gang_allreduce(x: float) -> float:
    import torch.distributed as dist

    from lzy_amd.runtime.context import op_context

    ctx = op_context()
    assert ctx is not None, "op_context missing inside gang op"
    assert ctx.gang_size == 2
    t = torch.tensor([x])
    dist.all_reduce(t, group=ctx.process_group)
    return float(t.item())


@op
def boom(x: int) -> int:
    raise ValueError("deliberate pool failure")


@whiteboard_("pool_wb")
@dataclass
class PoolWb:
    total: float
    label: str


def post_exit_access() -> None:
    """Results never touched inside the block stay readable after exit
    (reference tutorial 3: print(data_set.DESCR) outside the workflow)."""
    lzy = Lzy(runtime=GpuPoolRuntime())
    with lzy.workflow("post-exit-wf"):
        t = make_tensor(4096)
        s = inc(41)  # also never touched inside
    # touching AFTER the workflow: served from the durable tier
    assert int(s) == 42, int(s)
    assert float(t.sum().item()) == float(torch.arange(4096, dtype=torch.float32).sum())


def container_of_proxies() -> None:
    lzy = Lzy(runtime=GpuPoolRuntime())
    with lzy.workflow("pool-seq-wf"):
        parts = [inc(i) for i in range(4)]
        t = make_tensor(8)
        s = tensor_sum(t)
        total = sum(int(p) for p in parts) + int(float(s))
        assert total == (1 + 2 + 3 + 4) + 28, total


def client_abort() -> None:
    from lzy_amd.exceptions import WorkflowAbortedError

    lzy = Lzy(runtime=GpuPoolRuntime())
    try:
        with lzy.workflow("pool-abort-wf") as wf:
            inc(1)  # queued, dropped by abort
            wf.abort("operator said stop")
        raise AssertionError("expected WorkflowAbortedError")
    except WorkflowAbortedError as e:
        assert "operator said stop" in str(e)
    # the pool survives an abort: next workflow runs normally
    with lzy.workflow("pool-after-abort"):
        assert int(inc(10)) == 11


def main() -> None:
    # workers enter the serve loop here and never return
    GpuPool.get()

    lzy = Lzy(runtime=GpuPoolRuntime())

    # -- workflow 1: DAG with transfers -------------------------------------
    with lzy.workflow("pool-wf1") as wf:
        a = inc(1)
        b = inc(a)
        t = make_tensor(1024)
        s = tensor_sum(t)
        assert int(b) == 3, f"chain result {int(b)}"
        assert float(s) == float(torch.arange(1024, dtype=torch.float32).sum())

    # -- workflow 2: fan-out across ranks -----------------------------------
    with lzy.workflow("pool-wf2") as wf:
        ranks = [which_rank(i) for i in range(8)]
        seen = {int(r) for r in ranks}
        assert seen == {0, 1}, f"fan-out did not use both ranks: {seen}"

    # -- workflow 3: gang op with in-op collective --------------------------
    with lzy.workflow("pool-wf3") as wf:
        r = gang_allreduce(2.5)
        assert float(r) == 5.0, f"gang allreduce {float(r)}"

    # -- workflow 4: failure transport --------------------------------------
    try:
        with lzy.workflow("pool-wf4") as wf:
            y = boom(1)
            int(y)
        raise AssertionError("expected LzyExecutionError")
    except LzyExecutionError as e:
        assert "deliberate pool failure" in str(e)

    # -- workflow 5: whiteboard over the pool -------------------------------
    with lzy.workflow("pool-wf5") as wf:
        wb = wf.create_whiteboard(PoolWb, tags=["pool"])
        t = make_tensor(16)
        wb.total = tensor_sum(t)
        wb.label = "done"
        wb_id = wb.id
    got = lzy.whiteboard(id_=wb_id)
    assert got.total == 120.0
    assert got.label == "done"

    post_exit_access()
    container_of_proxies()
    client_abort()

    print("POOL-SCRIPT-OK", flush=True)


if __name__ == "__main__":
    main()
