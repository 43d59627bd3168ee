"""Chain-dispatch scenario (world 2, gloo): a same-rank dependency chain
is dispatched eagerly (no driver round-trip between links), results stay
correct, and a failing producer poisons its chained consumer promptly.
Prints CHAIN-OK on rank 0."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from lzy_amd import Lzy, op
from lzy_amd.exceptions import LzyExecutionError
from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime
from lzy_amd.utils.metrics import METRICS


@op
def slow_make(n: int) -> torch.Tensor:
    time.sleep(0.3)  # keep the chain inflight so children chain-dispatch
    return torch.arange(n, dtype=torch.float32)


@op
def double(t: torch.Tensor) -> torch.Tensor:
    return t * 2


@op
def total(t: torch.Tensor) -> float:
    return float(t.sum().item())


@op
def slow_boom(x: int) -> int:
    time.sleep(0.3)
    raise ValueError("chained producer failure")


@op
def consumer(x: int) -> int:
    return x + 1


def main() -> None:
    GpuPool.get()
    lzy = Lzy(runtime=GpuPoolRuntime())

    with lzy.workflow("chain-wf"):
        t = slow_make(512)
        s = total(double(double(t)))
        assert float(s) == float(torch.arange(512).sum()) * 4, float(s)
    chains = METRICS.counter_value("lzy_chain_dispatches")
    if os.environ.get("LZY_CHAIN_DISPATCH", "1") not in ("0", "false"):
        assert chains >= 2, f"expected chained dispatches, got {chains}"

    t0 = time.perf_counter()
    try:
        with lzy.workflow("chain-fail-wf"):
            # A(fails) -> B -> C: the whole chained suffix is poisoned
            y = consumer(consumer(slow_boom(1)))
            int(y)
        raise AssertionError("expected failure")
    except LzyExecutionError as e:
        assert "chained producer failure" in str(e), str(e)
    dt = time.perf_counter() - t0
    assert dt < 30, f"poisoned chain took {dt}s (timeout path?)"

    print("CHAIN-OK", flush=True)


if __name__ == "__main__":
    main()
