"""Spill-tier bookkeeping tests (CPU: injected pressure counters and a
fake mover; the real pinned-host DMA path is tests/test_gpu_workflow.py).
"""
import torch

from lzy_amd.storage.spill import SpillManager


class FakeCudaTensor:
    """Duck-typed 'device' tensor for bookkeeping tests."""

    def __init__(self, n):
        self._n = n
        self.is_cuda = True
        self.device = "fake"

    def numel(self):
        return self._n

    def element_size(self):
        return 4


class FakeMover:
    def __init__(self):
        self.spilled = []
        self.restored = []

    def to_host(self, t, entry_id=""):
        self.spilled.append(t)
        return ("host", t), None

    def to_device(self, host, device):
        self.restored.append(host)
        return host[1]


def _mgr(pressure, capacity=100, frac=0.5, mover=None):
    return SpillManager(
        device=None,
        threshold_frac=frac,
        bytes_in_use=pressure,
        capacity=lambda: capacity,
        mover=mover or FakeMover(),
    )


def test_no_spill_below_threshold():
    m = _mgr(lambda: 10)
    vals = {"a": FakeCudaTensor(4)}
    m.track("a", vals["a"])
    assert m.maybe_spill(vals) == 0
    assert m.spilled_count == 0


def test_spills_lru_first_until_below():
    state = {"use": 80}
    mover = FakeMover()

    def pressure():
        return state["use"]

    m = _mgr(pressure, mover=mover)
    vals = {k: FakeCudaTensor(4) for k in ["a", "b", "c"]}
    for k in ["a", "b", "c"]:
        m.track(k, vals[k])
    m.track("a", vals["a"])  # touch a -> LRU order is b, c, a

    orig = dict(vals)
    # each spill drops use by 30
    real_to_host = mover.to_host

    def to_host(t, entry_id=""):
        state["use"] -= 30
        return real_to_host(t, entry_id)

    mover.to_host = to_host
    n = m.maybe_spill(vals)
    assert n == 1
    assert m.is_spilled("b")            # least recently used went first
    assert not m.is_spilled("a")
    assert vals["b"] == ("host", orig["b"])


def test_unspill_restores_and_retracks():
    state = {"use": 80}
    m = _mgr(lambda: state["use"])
    vals = {"x": FakeCudaTensor(4)}
    m.track("x", vals["x"])
    orig = vals["x"]

    def drop(t, entry_id=""):
        state["use"] = 0
        return ("host", t), None

    m._mover.to_host = drop
    assert m.maybe_spill(vals) == 1
    assert m.is_spilled("x")
    back = m.unspill("x", vals)
    assert back is orig
    assert vals["x"] is orig
    assert not m.is_spilled("x")


def test_forget_clears_both_tiers():
    m = _mgr(lambda: 0)
    v = FakeCudaTensor(1)
    m.track("e", v)
    m.forget("e")
    assert m.maybe_spill({"e": v}) == 0


def test_host_tensors_ignored():
    m = _mgr(lambda: 99)
    vals = {"h": torch.ones(4)}  # plain cpu tensor
    m.track("h", vals["h"])
    assert m.maybe_spill(vals) == 0
