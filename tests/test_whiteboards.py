"""Whiteboard lifecycle (reference scenario whiteboards + WhiteboardService
tests): declare, create, assign op outputs, FINALIZE on exit, read back by
id, query by name/tags/time."""
import datetime
from dataclasses import dataclass, field

import pytest

from lzy_amd import Lzy, op, whiteboard_


@whiteboard_("train_result")
@dataclass
class TrainResult:
    accuracy: float
    epochs: int
    note: str = "default-note"


@op
def compute_acc(x: float) -> float:
    return x * 2


def test_whiteboard_roundtrip(lzy):
    with lzy.workflow("wf") as wf:
        wb = wf.create_whiteboard(TrainResult, tags=["exp1", "v1"])
        wb.accuracy = compute_acc(0.42)
        wb.epochs = 10
        wb_id = wb.id

    got = lzy.whiteboard(id_=wb_id)
    assert got is not None
    assert got.status == "FINALIZED"
    assert got.name == "train_result"
    assert got.accuracy == 0.84
    assert got.epochs == 10
    assert got.note == "default-note"


def test_whiteboard_query(lzy):
    with lzy.workflow("wf") as wf:
        wb = wf.create_whiteboard(TrainResult, tags=["tag_a", "tag_b"])
        wb.accuracy = 0.5
        wb.epochs = 1

    boards = list(lzy.whiteboards(name="train_result", tags=["tag_a"]))
    assert len(boards) == 1
    assert boards[0].accuracy == 0.5

    assert list(lzy.whiteboards(name="nope")) == []
    assert list(lzy.whiteboards(name="train_result", tags=["other"])) == []

    future = datetime.datetime.now(datetime.timezone.utc) + datetime.timedelta(days=1)
    assert list(lzy.whiteboards(name="train_result", not_before=future)) == []


def test_unassigned_field_fails(lzy):
    with pytest.raises(RuntimeError, match="never assigned"):
        with lzy.workflow("wf") as wf:
            wb = wf.create_whiteboard(TrainResult)
            wb.accuracy = 0.1
            # epochs never assigned -> finalize at exit fails


def test_non_whiteboard_class_rejected(lzy):
    @dataclass
    class Plain:
        x: int

    with lzy.workflow("wf") as wf:
        with pytest.raises(TypeError):
            wf.create_whiteboard(Plain)
        # drain: nothing queued


def test_unstable_field_type_rejected(lzy):
    class Weird:
        pass

    @whiteboard_("weird_wb")
    @dataclass
    class WeirdWb:
        data: Weird = None

    with lzy.workflow("wf") as wf:
        with pytest.raises(TypeError, match="stable"):
            wf.create_whiteboard(WeirdWb)


def test_whiteboard_name_validation():
    with pytest.raises(ValueError):
        whiteboard_("bad name!")(None)
    with pytest.raises(TypeError):
        whiteboard_("")(None)


def test_whiteboard_tensor_field(lzy):
    import torch

    @whiteboard_("model_wb")
    @dataclass
    class ModelWb:
        weights: torch.Tensor = None

    with lzy.workflow("wf") as wf:
        wb = wf.create_whiteboard(ModelWb)
        wb.weights = torch.arange(16, dtype=torch.float32).reshape(4, 4)
        wb_id = wb.id

    got = lzy.whiteboard(id_=wb_id)
    assert torch.equal(got.weights, torch.arange(16, dtype=torch.float32).reshape(4, 4))
