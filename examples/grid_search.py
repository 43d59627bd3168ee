"""Grid-search example: fan-out across the node's GPUs, gather into a
whiteboard (BASELINE config 3 shape).

Run single-process (CPU ok):     python examples/grid_search.py
Run one process per GPU:         python -m torch.distributed.run \
    --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 examples/grid_search.py
"""
import json
import os
import sys
from dataclasses import dataclass

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from lzy_amd import Lzy, op, whiteboard_


@whiteboard_("grid_search_result")
@dataclass
class GridResult:
    # whiteboard fields need STABLE serializers (same rule as the
    # reference): containers travel as JSON text
    best_lr: float = 0.0
    best_loss: float = 0.0
    all_scores_json: str = "{}"



@op
def train_candidate(lr: float, steps: int) -> float:
    """One hyper-parameter candidate: tiny regression fit; placed on
    whichever rank/GPU the scheduler picks."""
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    torch.manual_seed(0)
    w_true = torch.randn(16, 1, device=dev)
    x = torch.randn(512, 16, device=dev)
    y = x @ w_true
    model = torch.nn.Linear(16, 1, bias=False, device=dev)
    opt = torch.optim.SGD(model.parameters(), lr=lr)
    for _ in range(steps):
        opt.zero_grad(set_to_none=True)
        loss = (model(x) - y).square().mean()
        loss.backward()
        opt.step()
    return float(loss.item())


def main() -> None:
    lzy = Lzy()
    lrs = [3e-4, 1e-3, 3e-3, 1e-2, 3e-2, 1e-1]
    with lzy.workflow("grid-search", interactive=False) as wf:
        wb = wf.create_whiteboard(GridResult, tags=["example", "grid"])
        losses = {lr: train_candidate(lr, steps=50) for lr in lrs}
        scores = {str(lr): float(l) for lr, l in losses.items()}
        best_lr = min(scores, key=scores.get)
        wb.best_lr = float(best_lr)
        wb.best_loss = scores[best_lr]
        wb.all_scores_json = json.dumps(scores, sort_keys=True)
        wb_id = wb.id

    back = lzy.whiteboard(id_=wb_id)
    scores = json.loads(back.all_scores_json)
    print(f"best lr={back.best_lr} loss={back.best_loss:.6f} "
          f"({len(scores)} candidates)")
    print("GRID-EXAMPLE-OK")


if __name__ == "__main__":
    main()
