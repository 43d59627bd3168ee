"""Map-reduce example with a chunk-streamed merge tree (BASELINE
config-4 shape: large tensor slots between ops).

``@op(pair_reduce=(alpha, beta))`` declares the op's semantics —
``alpha*a + beta*b`` over two same-shape tensors — which lets the pool
runtime fold the whole reduction tree into ONE chunk-pipelined transfer
plan over xGMI instead of log2(N) full-shard transfer-then-combine
levels.  The function body is still what runs on the LocalRuntime, on
same-rank edges, and whenever folding is not applicable — declare only
what the body actually computes.

Run single-process (CPU ok):     python examples/map_reduce_stream.py
Run one process per GPU:         python -m torch.distributed.run \
    --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
    examples/map_reduce_stream.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from lzy_amd import Lzy, op

SHARD = 1 << 22  # 16 MiB f32 per shard (keep the example quick)


def _device() -> torch.device:
    if torch.cuda.is_available():
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


@op
def load_shard(i: int) -> torch.Tensor:
    g = torch.Generator().manual_seed(42 + i)
    t = torch.randn(SHARD, generator=g)
    return t.to(_device()) if _device().type == "cuda" else t


@op
def featurize(t: torch.Tensor) -> torch.Tensor:
    return torch.tanh(t) * 2.0


@op(pair_reduce=(0.5, 0.5))
def mean2(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    # exactly what pair_reduce declares: 0.5*a + 0.5*b
    return (a + b) * 0.5


@op
def summarize(t: torch.Tensor) -> float:
    return float(t.float().abs().mean())


def main() -> None:
    from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime

    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        GpuPool.get()  # workers enter the serve loop here
        lzy = Lzy(runtime=GpuPoolRuntime())
    else:
        lzy = Lzy()  # LocalRuntime

    width = max(2, world)
    with lzy.workflow("map-reduce-stream", interactive=False):
        shards = [featurize(load_shard(i)) for i in range(width)]
        layer = shards
        while len(layer) > 1:
            layer = [
                mean2(layer[i], layer[i + 1])
                for i in range(0, len(layer) - 1, 2)
            ] + ([layer[-1]] if len(layer) % 2 else [])
        score = summarize(layer[0])
        print(f"mean |feature| over {width} shards: {float(score):.5f}")

    from lzy_amd.utils.metrics import METRICS

    plans = METRICS.counter_value("lzy_stream_plans")
    if world > 1:
        print(f"streamed merge plans used: {plans}")


if __name__ == "__main__":
    main()
