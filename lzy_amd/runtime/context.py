"""Op execution context for gang-scheduled (multi-GPU) ops.

Reference analogue: the reference only *guards* user-launched DDP inside
an op (LZY_OP_MAIN_PID, pylzy/lzy/api/v1/startup.py:80-106).  Here
multi-GPU ops are first-class: ``@op(gpu_count=k)`` gang-schedules the
function onto k ranks, and inside the op ``op_context()`` exposes the
gang rank/size and a ready RCCL process group for collectives/DDP.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Optional, Tuple


@dataclass
class OpContext:
    gang_rank: int
    gang_size: int
    ranks: Tuple[int, ...]
    process_group: object = None  # torch.distributed.ProcessGroup

    @property
    def is_primary(self) -> bool:
        return self.gang_rank == 0

    @property
    def device(self):
        import torch

        if torch.cuda.is_available():
            return torch.device("cuda", torch.cuda.current_device())
        return torch.device("cpu")


_ctx = threading.local()


def _set_op_context(ctx: Optional[OpContext]) -> None:
    _ctx.value = ctx


def op_context() -> Optional[OpContext]:
    """The gang context of the currently executing op (None outside gangs)."""
    return getattr(_ctx, "value", None)


_in_task = threading.local()


def _set_in_op_execution(flag: bool) -> None:
    _in_task.value = flag


def in_op_execution() -> bool:
    """True on a thread currently executing an op body.  Nested
    ``Lzy().workflow()`` started from inside an op must run in-process
    (LocalRuntime) — a pool worker cannot become a pool driver
    (reference analogue: nested graphs run through the op's own client,
    scenarios/nested_workflows)."""
    return bool(getattr(_in_task, "value", False))
