"""Gang-scheduled data-parallel training example (BASELINE config 5
shape): a single ``@op(gpu_count=N)`` runs on N GPUs simultaneously with
an RCCL process group for gradient all-reduce.

Run:  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
          --master-addr 127.0.0.1 examples/train_ddp.py
(CPU/gloo works too for a smoke: --nproc-per-node 2.)
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from lzy_amd import Lzy, op

WORLD = int(os.environ.get("WORLD_SIZE", "1"))


@op(gpu_count=max(1, WORLD))
def train_epoch(dim: int, batch: int, steps: int) -> float:
    import torch.distributed as dist

    from lzy_amd.runtime.context import op_context

    ctx = op_context()
    # one rank per GPU is the production shape; in oversubscribed
    # harnesses (more ranks than GPUs) RCCL refuses two ranks on one
    # device, so collectives fall back to CPU/gloo there
    world = len(ctx.ranks) if ctx is not None else 1
    use_cuda = torch.cuda.is_available() and torch.cuda.device_count() >= world
    dev = torch.device("cuda") if use_cuda else torch.device("cpu")
    torch.manual_seed(1234 + (ctx.gang_rank if ctx else 0))

    model = torch.nn.Sequential(
        torch.nn.Linear(dim, dim, device=dev),
        torch.nn.GELU(),
        torch.nn.Linear(dim, 10, device=dev),
    )
    # replicas start identical: broadcast rank-0 weights
    if ctx is not None and ctx.gang_size > 1:
        for p in model.parameters():
            dist.broadcast(p.data, src=0, group=ctx.process_group)
    opt = torch.optim.SGD(model.parameters(), lr=1e-2)

    loss_val = 0.0
    for _ in range(steps):
        x = torch.randn(batch, dim, device=dev)
        y = torch.randint(0, 10, (batch,), device=dev)
        opt.zero_grad(set_to_none=True)
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        if ctx is not None and ctx.gang_size > 1:
            for p in model.parameters():  # DDP-style grad all-reduce
                dist.all_reduce(p.grad, group=ctx.process_group)
                p.grad /= ctx.gang_size
        opt.step()
        loss_val = float(loss.item())
    return loss_val


def main() -> None:
    lzy = Lzy()
    with lzy.workflow("ddp-train", interactive=False):
        final = train_epoch(dim=256, batch=64, steps=5)
        print(f"final loss {float(final):.4f}")
    print("DDP-EXAMPLE-OK")


if __name__ == "__main__":
    main()
