"""Map-reduce with pair_reduce merge ops: identical output whether the
tree runs op-by-op (LocalRuntime / fallback) or as a chunk-streamed plan
(pool engine at world > 1) — the analogue of the reference's
large_input_output scenario for the round-2 streamed data plane."""
import torch

from lzy_amd import Lzy, op

N = 1 << 18  # 1 MiB f32 shards


@op
def shard(i: int) -> torch.Tensor:
    g = torch.Generator().manual_seed(500 + i)
    return torch.randn(N, generator=g)


@op(pair_reduce=(0.5, 0.5))
def mean2(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return (a + b) * 0.5


@op
def digest(t: torch.Tensor) -> str:
    return f"sum={float(t.sum()):.4f} absmax={float(t.abs().max()):.4f}"


if __name__ == "__main__":
    lzy = Lzy()
    with lzy.workflow("stream_merge", interactive=False):
        layer = [shard(i) for i in range(4)]
        m01 = mean2(layer[0], layer[1])
        m23 = mean2(layer[2], layer[3])
        root = mean2(m01, m23)
        print(str(digest(root)))
        print(str(digest(m01)))  # interior value stays readable
    print("DONE")
