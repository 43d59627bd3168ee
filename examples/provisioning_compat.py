"""Reference provisioning example, unchanged idioms (cf.
pylzy/examples/provisioning.py): env shortcuts stacked as decorators
above @op, the Any sentinel, and reference-named env classes — all
running against the MI355X pool (or LocalRuntime on CPU).

Run:  python examples/provisioning_compat.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from lzy_amd import AnyProvisioning, Lzy, op
from lzy_amd.env.shortcuts import env_vars, gpu, provisioning


@op
def example1() -> int:
    return 1


@provisioning(cpu_count=8)
@op
def example2() -> int:
    return 2


@provisioning(gpu_count=AnyProvisioning)
@op
def example3() -> int:
    return 3


@gpu(1)
@op
def example4() -> int:
    import torch

    return 4 + int(torch.zeros(1).sum())


@env_vars(EXAMPLE_FLAVOR="mi355x")
@op
def example5() -> str:
    import os

    return os.environ["EXAMPLE_FLAVOR"]


def main() -> None:
    lzy = Lzy()
    with lzy.workflow("provisioning-compat"):
        total = int(example1()) + int(example2()) + int(example3())
        flavor = str(example5())
    print(f"sum={total} flavor={flavor}")
    assert total == 6 and flavor == "mi355x"


if __name__ == "__main__":
    main()
