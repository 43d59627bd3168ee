"""Whiteboard write/finalize/read-back by id and by query (reference
scenario: whiteboards)."""
from dataclasses import dataclass

from lzy_amd import Lzy, op, whiteboard_


@whiteboard_("scenario_wb")
@dataclass
class Result:
    score: float = 0.0
    label: str = "none"


@op
def compute(x: float) -> float:
    return x * 2.0


if __name__ == "__main__":
    lzy = Lzy()
    with lzy.workflow("wf", interactive=False) as wf:
        wb = wf.create_whiteboard(Result, tags=["scenario", "v1"])
        wb.score = compute(21.0)
        wb.label = "trained"
        wb_id = wb.id

    back = lzy.whiteboard(id_=wb_id)
    print(back.score)
    print(back.label)
    print(back.status)

    found = list(lzy.whiteboards(name="scenario_wb", tags=["scenario"]))
    print(len(found) >= 1)
