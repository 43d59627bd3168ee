"""End-to-end workflow on one MI355X: device tensors through the DAG,
HBM-resident store, GPU train op, result cache for device tensors."""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")


@requires_gpu
def test_device_tensor_dag(lzy):
    from lzy_amd import op

    @op
    def make(n: int) -> torch.Tensor:
        from lzy_amd.ops import fill_pattern

        t = torch.empty(n, device="cuda", dtype=torch.bfloat16)
        fill_pattern(t, seed=1)
        return t

    @op
    def scale(t: torch.Tensor) -> torch.Tensor:
        return t * 2

    @op
    def reduce_sum(t: torch.Tensor) -> float:
        return float(t.float().sum().item())

    with lzy.workflow("gpu-wf") as wf:
        t = make(1 << 20)
        s = scale(t)
        total = reduce_sum(s)
        v = float(total)
    assert v != 0.0


@requires_gpu
def test_gpu_train_op(lzy):
    from lzy_amd import op

    @op
    def train(n: int) -> float:
        x = torch.randn(n, 256, device="cuda", dtype=torch.bfloat16)
        model = torch.nn.Linear(256, 64).to("cuda", torch.bfloat16)
        opt = torch.optim.AdamW(model.parameters())
        loss = model(x).float().square().mean()
        loss.backward()
        opt.step()
        return float(loss.item())

    with lzy.workflow("train-wf") as wf:
        l = train(1024)
        assert float(l) > 0


@requires_gpu
def test_cache_device_tensor(lzy):
    from lzy_amd import op

    runs = []

    @op(cache=True, version="1.0")
    def expensive(n: int) -> torch.Tensor:
        runs.append(n)
        return torch.full((n,), 3.0, device="cuda")

    with lzy.workflow("c1"):
        a = expensive(4096)
        assert float(a.float().sum().item()) == 3.0 * 4096
    with lzy.workflow("c2"):
        b = expensive(4096)
        # loaded from cache -> tensor comes back (CPU or GPU), same content
        assert float(b.float().sum().item()) == 3.0 * 4096
    assert runs == [4096]


@requires_gpu
def test_pool_runtime_single_gpu(tmp_path, monkeypatch):
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    env = {
        **__import__("os").environ,
        "LZY_AMD_STORAGE": str(tmp_path / "s"),
        "PYTHONPATH": str(root),
    }
    res = subprocess.run(
        [sys.executable, "tests/pool_script_single.py"],
        cwd=root, env=env, capture_output=True, text=True, timeout=300,
    )
    assert res.returncode == 0, res.stderr[-2000:]
    assert "SINGLE-OK" in res.stdout


@requires_gpu
def test_per_op_streams_overlap_and_events(lzy):
    """Fan-out ops land on distinct HIP streams; device hand-off is
    event-ordered (runtime/streams.py) and results stay correct."""
    from lzy_amd import op
    from lzy_amd.runtime.streams import STREAMS
    from lzy_amd.utils.metrics import METRICS

    base_rec = METRICS.counter_value("lzy_stream_events_recorded")

    @op
    def make(i: int) -> torch.Tensor:
        return torch.full((1 << 20,), float(i + 1), device="cuda")

    @op
    def double(t: torch.Tensor) -> torch.Tensor:
        return t * 2

    @op
    def s(t: torch.Tensor) -> float:
        return float(t.float().sum().item())

    with lzy.workflow("streams-wf"):
        outs = [s(double(make(i))) for i in range(6)]
        got = [float(o) for o in outs]
    assert got == [2.0 * (i + 1) * (1 << 20) for i in range(6)]
    assert METRICS.counter_value("lzy_stream_events_recorded") > base_rec


@requires_gpu
def test_stream_race_checker_detects_unordered_put(monkeypatch):
    """LZY_STREAM_CHECK=1: a device tensor that entered the store without
    a recorded producing event is flagged as a race at read time."""
    from lzy_amd.runtime.streams import STREAMS

    monkeypatch.setenv("LZY_STREAM_CHECK", "1")
    t = torch.ones(16, device="cuda")
    with pytest.raises(RuntimeError, match="stream-race"):
        STREAMS.wait_value("never-recorded-entry", t)


@requires_gpu
def test_pinned_host_spill_roundtrip():
    """HBM store spill tier: past the (test-lowered) threshold, LRU
    tensors move to pinned host and come back bit-identical on access."""
    from lzy_amd.runtime.taskspec import WorkerStore
    from lzy_amd.storage.spill import SpillManager, _DefaultMover

    dev = torch.device("cuda", 0)
    store = WorkerStore(device=dev)
    # force pressure: everything above 0 bytes spills
    store.spill = SpillManager(
        device=dev,
        threshold_frac=0.0,
        bytes_in_use=lambda: 1,
        capacity=lambda: 1,
        mover=_DefaultMover(dev),
    )
    a = torch.randn(1 << 20, device=dev)
    b = torch.randn(1 << 20, device=dev)
    ref_a, ref_b = a.cpu().clone(), b.cpu().clone()
    store.put("a", a)
    assert store.spill.is_spilled("a")
    host_a = store.values["a"]
    assert not host_a.is_cuda and host_a.is_pinned()
    store.put("b", b)
    assert store.spill.is_spilled("b")

    back_a = store.get("a")
    assert back_a.is_cuda
    assert torch.equal(back_a.cpu(), ref_a)
    back_b = store.get("b")
    assert torch.equal(back_b.cpu(), ref_b)
    assert not store.spill.is_spilled("a")


@requires_gpu
def test_scenarios_on_gpu(tmp_path):
    """A slice of the e2e scenario suite executed on the GPU box (ops
    allocate device tensors through the full framework stack)."""
    import re
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    for name in ["complex_graph", "whiteboards", "fully_cached_graph"]:
        scen = root / "tests" / "scenarios" / name
        env = {**__import__("os").environ,
               "LZY_AMD_STORAGE": str(tmp_path / name),
               "PYTHONPATH": str(root)}
        r = subprocess.run(
            [sys.executable, str(scen / "__init__.py")],
            cwd=root, env=env, capture_output=True, text=True, timeout=240,
        )
        assert r.returncode == 0, f"{name}: {r.stdout[-1500:]}{r.stderr[-1500:]}"
        expected = (scen / "expected_stdout").read_text().splitlines()
        actual = [re.sub(r"^\[LZY-[^\]]*\] ", "", l)
                  for l in r.stdout.splitlines() if l.strip()]
        for line in expected:
            if line != "...":
                assert line in actual, f"{name}: missing {line!r} in {actual}"


@requires_gpu
def test_stepgraph_captures_on_gpu():
    """hipGraph capture really engages on ROCm (no silent eager)."""
    from lzy_amd.utils.hipgraph import StepGraph

    model = torch.nn.Linear(64, 64, device="cuda")
    opt = torch.optim.SGD(model.parameters(), lr=0.01)

    def step(x):
        opt.zero_grad(set_to_none=True)
        loss = model(x).square().mean()
        loss.backward()
        opt.step()
        return loss.detach()

    sg = StepGraph(step)
    x = torch.randn(32, 64, device="cuda")
    l1 = float(sg.run(x).item())
    l2 = float(sg.run(torch.randn(32, 64, device="cuda")).item())
    assert sg.captured and not sg.fallback_eager
    assert l1 > 0 and l2 > 0 and l2 == l2  # finite
    # replay really trains: loss trends down over repeats of same input
    xs = torch.randn(32, 64, device="cuda")
    first = float(sg.run(xs).item())
    for _ in range(20):
        last = float(sg.run(xs).item())
    assert last < first


@requires_gpu
def test_record_if_absent_gpu_contract():
    from lzy_amd.runtime.streams import StreamPlacer

    p = StreamPlacer()
    t = torch.ones(8, device="cuda")
    s1 = torch.cuda.Stream()
    p.record_output("e", t, stream=s1)          # precise producer record
    ev1 = p._events["e"][0]
    p.record_output("e", t, if_absent=True)     # blanket hook: must keep
    assert p._events["e"][0] is ev1
    p.record_output("e", t)                     # explicit: may replace
    assert p._events["e"][0] is not ev1
