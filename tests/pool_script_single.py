"""ws=1 pool scenario (no torch.distributed init)."""
import torch

from lzy_amd import Lzy, op
from lzy_amd.runtime.pool import GpuPoolRuntime


@op
def double(t: torch.Tensor) -> torch.Tensor:
    return t * 2


@op
def total(t: torch.Tensor) -> float:
    return float(t.sum().item())


def main() -> None:
    lzy = Lzy(runtime=GpuPoolRuntime())
    with lzy.workflow("single") as wf:
        t = double(torch.ones(64))
        s = total(t)
        assert float(s) == 128.0
    print("SINGLE-OK", flush=True)


if __name__ == "__main__":
    main()
