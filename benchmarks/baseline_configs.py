"""The five BASELINE.json measurement configs, runnable individually.

  1  single @op identity fn via LocalRuntime on CPU (plumbing)
  2  2-stage preprocess -> GBDT/train DAG, gpu_count=1
  3  N-way fan-out grid-search @op map, whiteboard gather
  4  4-stage pipeline with 4 GB tensor slots between ops (xGMI channels)
  5  @op(gpu_count=N) DDP ResNet-50 step, RCCL all-reduce inside the op

Usage:  python benchmarks/baseline_configs.py --config 3 --iters 5
Multi-GPU configs scale to WORLD_SIZE when launched under torchrun
(one rank per GPU); at world 1 they degenerate but still execute.
Each config prints one JSON line: {"config": k, "makespan_s": ...}.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time
from dataclasses import dataclass

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from lzy_amd import Lzy, op, whiteboard_
from lzy_amd.runtime.local import LocalRuntime


def _gpu() -> bool:
    return torch.cuda.is_available()


# -- config 1 ----------------------------------------------------------------

@op
def identity(x: int) -> int:
    return x


def config1(lzy: Lzy, _: int) -> None:
    with lzy.workflow("c1"):
        assert int(identity(42)) == 42


# -- config 2: preprocess -> train, gpu_count=1 ------------------------------

@op
def c2_preprocess(n: int) -> torch.Tensor:
    dev = "cuda" if _gpu() else "cpu"
    x = torch.randn(n, 128, device=dev)
    return (x - x.mean(0)) / (x.std(0) + 1e-6)


@op(gpu_count=1)
def c2_train(x: torch.Tensor) -> float:
    # gradient-boosted-trees-style objective on GPU: iterative stumps via
    # torch histograms (catboost itself is not in the image); the op IS
    # placed on one MI355X like the reference's CatBoost GPU pool task
    y = (x[:, 0] * 0.5 + x[:, 1].sin()).detach()
    pred = torch.zeros_like(y)
    for _ in range(20):
        resid = y - pred
        f = x[:, int(torch.argmax(resid.abs()).item()) % x.shape[1]]
        thr = f.median()
        left = resid[f <= thr].mean() if (f <= thr).any() else 0.0
        right = resid[f > thr].mean() if (f > thr).any() else 0.0
        pred = pred + torch.where(f <= thr, left, right) * 0.3
    return float((y - pred).square().mean().item())


def config2(lzy: Lzy, _: int) -> None:
    with lzy.workflow("c2"):
        x = c2_preprocess(1 << 16 if _gpu() else 1 << 12)
        loss = c2_train(x)
        assert float(loss) >= 0


# -- config 3: fan-out grid search + whiteboard gather -----------------------

@whiteboard_("grid_search")
@dataclass
class GridResult:
    best_lr: float
    best_score: float


@op
def c3_eval(lr: float, n: int) -> float:
    dev = "cuda" if _gpu() else "cpu"
    x = torch.randn(n, 256, device=dev)
    w = torch.zeros(256, device=dev)
    target = torch.randn(256, device=dev)
    for _ in range(30):
        grad = (w - target) + 0.01 * torch.randn_like(w)
        w = w - lr * grad
    return float((w - target).square().mean().item())


def config3(lzy: Lzy, width: int) -> None:
    lrs = [0.02 * (i + 1) for i in range(width)]
    with lzy.workflow("c3") as wf:
        wb = wf.create_whiteboard(GridResult, tags=["grid"])
        scores = [c3_eval(lr, 1 << 16 if _gpu() else 1 << 10) for lr in lrs]
        vals = [float(s) for s in scores]
        best = min(range(len(vals)), key=lambda i: vals[i])
        wb.best_lr = lrs[best]
        wb.best_score = vals[best]


# -- config 4: 4-stage pipeline with 4 GB tensor slots -----------------------

C4_GB = float(os.environ.get("LZY_C4_GB", "4"))


@op
def c4_source(seed: int) -> torch.Tensor:
    n = int(C4_GB * (1 << 30) // 2)  # bf16
    dev = "cuda" if _gpu() else "cpu"
    t = torch.empty(n, dtype=torch.bfloat16, device=dev)
    if _gpu():
        from lzy_amd.ops import fill_pattern

        fill_pattern(t, seed=seed, mask16=0x3FFF)
    else:
        t.zero_()
    return t


@op
def c4_stage(t: torch.Tensor) -> torch.Tensor:
    return t * 1.001


@op
def c4_sink(t: torch.Tensor) -> float:
    return float(t[::65537].float().sum().item())


def config4(lzy: Lzy, _: int) -> None:
    with lzy.workflow("c4"):
        t = c4_source(1)
        a = c4_stage(t)
        b = c4_stage(a)
        s = c4_sink(b)
        assert float(s) == float(s)


# -- config 5: gang DDP ResNet-50 step ---------------------------------------

@op(gpu_count=int(os.environ.get("WORLD_SIZE", "1")))
def c5_resnet_step(batch: int) -> float:
    import torch.distributed as dist

    from lzy_amd.models import resnet50
    from lzy_amd.runtime.context import op_context

    dev = torch.device("cuda") if _gpu() else torch.device("cpu")
    global _C5_STATE
    try:
        model, opt = _C5_STATE
    except NameError:
        model = resnet50(num_classes=1000).to(dev)
        opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
        _C5_STATE = (model, opt)

    x = torch.randn(batch, 3, 224, 224, device=dev)
    y = torch.randint(0, 1000, (batch,), device=dev)
    opt.zero_grad(set_to_none=True)
    loss = torch.nn.functional.cross_entropy(model(x), y)
    loss.backward()

    ctx = op_context()
    if ctx is not None and ctx.gang_size > 1:
        # bucketless hand all-reduce of grads over RCCL/xGMI
        for p in model.parameters():
            if p.grad is not None:
                dist.all_reduce(p.grad, group=ctx.process_group)
                p.grad /= ctx.gang_size
    opt.step()
    return float(loss.detach().item())


def config5(lzy: Lzy, _: int) -> None:
    with lzy.workflow("c5"):
        loss = c5_resnet_step(32 if _gpu() else 2)
        assert float(loss) > 0


CONFIGS = {1: config1, 2: config2, 3: config3, 4: config4, 5: config5}


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", type=int, required=True, choices=sorted(CONFIGS))
    ap.add_argument("--iters", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    if args.config == 1 and world == 1:
        lzy = Lzy(runtime=LocalRuntime())
        pool = None
    else:
        from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime

        pool = GpuPool.get()
        lzy = Lzy(runtime=GpuPoolRuntime())

    fn = CONFIGS[args.config]
    width = max(world, 1) if args.config == 3 else world
    if args.config == 3 and world == 1:
        width = 8  # grid of 8 points even on one GPU

    for i in range(args.warmup):
        fn(lzy, width)
    if pool is not None:
        pool.sync_all()
    t0 = time.perf_counter()
    for i in range(args.iters):
        fn(lzy, width)
    if pool is not None:
        pool.sync_all()
    dt = (time.perf_counter() - t0) / args.iters
    print(json.dumps({
        "config": args.config,
        "makespan_s": round(dt, 6),
        "iters": args.iters,
        "world": world,
        "gpu": _gpu(),
    }), flush=True)


if __name__ == "__main__":
    main()
