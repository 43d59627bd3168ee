"""Large (multi-hundred-MB) tensors through the data plane (reference
scenario: large_input_output — 'large IO' e2e)."""
import torch

from lzy_amd import Lzy, op


@op
def big(n: int) -> torch.Tensor:
    return torch.full((n,), 3.0, dtype=torch.float32)


@op
def reduce_it(t: torch.Tensor) -> float:
    return float(t.sum().item())


if __name__ == "__main__":
    n = 64 << 20  # 256 MB f32
    with Lzy().workflow("wf", interactive=False):
        s = reduce_it(big(n))
        print(int(float(s)) == 3 * n)
    print("large io ok")
