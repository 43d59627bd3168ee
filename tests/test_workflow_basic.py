"""Phase-0 workflow semantics (reference scenarios: complex_graph, exec_fail).

Covers: @op capture, laziness, barrier, multi-output, fan-out/fan-in DAG,
exception transport, ops outside workflows.
"""
import pytest

from lzy_amd import Lzy, op, materialize, is_lzy_proxy
from lzy_amd.exceptions import LzyExecutionError


@op
def inc(x: int) -> int:
    return x + 1


@op
def add(a: int, b: int) -> int:
    return a + b


@op
def split(x: int) -> (int, int):
    return x, x + 10


def test_single_identity_op(lzy):
    @op
    def ident(x: int) -> int:
        return x

    with lzy.workflow("wf") as wf:
        y = ident(42)
        assert is_lzy_proxy(y)
        assert int(y) == 42


def test_chain(lzy):
    with lzy.workflow("wf") as wf:
        a = inc(1)
        b = inc(a)
        c = inc(b)
        assert int(c) == 4


def test_fan_out_fan_in(lzy):
    with lzy.workflow("wf") as wf:
        xs = [inc(i) for i in range(8)]
        total = add(add(xs[0], xs[1]), add(xs[2], xs[3]))
        assert int(total) == (1 + 2 + 3 + 4)


def test_multi_output(lzy):
    with lzy.workflow("wf") as wf:
        lo, hi = split(5)
        assert int(lo) == 5
        assert int(hi) == 15


def test_tuple_annotation(lzy):
    from typing import Tuple

    @op
    def pair(x: int) -> Tuple[int, str]:
        return x, str(x)

    with lzy.workflow("wf"):
        a, b = pair(7)
        assert int(a) == 7
        assert str(b) == "7"


def test_materialize_after_exit(lzy):
    with lzy.workflow("wf") as wf:
        y = inc(10)
    # proxy still materializable after workflow end
    assert int(y) == 11


def test_op_outside_workflow_runs_directly():
    assert inc(5) == 6


def test_missing_return_annotation():
    with pytest.raises(TypeError):
        @op
        def bad(x):
            return x


def test_exception_transport(lzy):
    @op
    def boom(x: int) -> int:
        raise ValueError("broken op")

    with pytest.raises(LzyExecutionError) as ei:
        with lzy.workflow("wf"):
            y = boom(1)
            int(y)  # force barrier
    assert "broken op" in str(ei.value)
    assert "ValueError" in ei.value.remote_traceback


def test_downstream_cancelled_on_failure(lzy):
    ran = []

    @op
    def fail(x: int) -> int:
        raise RuntimeError("nope")

    @op
    def after(x: int) -> int:
        ran.append(x)
        return x

    with pytest.raises(LzyExecutionError):
        with lzy.workflow("wf"):
            y = fail(1)
            z = after(y)
            int(z)
    assert ran == []


def test_eager_mode(lzy):
    order = []

    @op
    def track(x: int) -> int:
        order.append(x)
        return x

    with lzy.workflow("wf", eager=True):
        track(1)
        assert order == [1]
        track(2)
        assert order == [1, 2]


def test_none_output(lzy):
    @op
    def nothing(x: int) -> None:
        return None

    with lzy.workflow("wf"):
        r = nothing(1)
        assert materialize(r) is None


def test_proxy_kwargs(lzy):
    @op
    def combine(a: int, *, b: int = 0) -> int:
        return a * 100 + b

    with lzy.workflow("wf"):
        x = inc(1)
        y = combine(3, b=x)
        assert int(y) == 302


def test_independent_ops_parallel(lzy):
    import threading
    import time

    barrier = threading.Barrier(4, timeout=10)

    @op
    def wait_all(i: int) -> int:
        barrier.wait()
        return i

    with lzy.workflow("wf"):
        rs = [wait_all(i) for i in range(4)]
        assert sorted(int(r) for r in rs) == [0, 1, 2, 3]


def test_lazy_arguments_defer_materialization(lzy):
    """@op(lazy_arguments=True): the op body receives proxies and decides
    what to touch (reference: startup.py lazy slot reads)."""
    from lzy_amd.proxy import is_lzy_proxy

    seen = {}

    @op
    def src(x: int) -> int:
        return x * 2

    @op(lazy_arguments=True)
    def probe(a: int, b: int) -> int:
        seen["a_proxy"] = is_lzy_proxy(a)
        seen["b_proxy"] = is_lzy_proxy(b)
        return int(a) + 1  # touch only a

    with lzy.workflow("wf", interactive=False):
        r = probe(src(5), src(7))
        assert int(r) == 11
    assert seen["a_proxy"] and seen["b_proxy"]


def test_interactive_confirm(lzy, monkeypatch):
    """interactive=True on a TTY prompts before the first submission;
    declining aborts the workflow (reference runtime.py:424)."""
    import sys

    from lzy_amd.exceptions import WorkflowAbortedError

    @op
    def f(x: int) -> int:
        return x + 1

    class FakeTty:
        def isatty(self):
            return True

    monkeypatch.setattr(sys, "stdin", FakeTty())
    prompts = []

    monkeypatch.setattr("builtins.input", lambda msg: (prompts.append(msg), "y")[1])
    with lzy.workflow("ok-wf", interactive=True):
        assert int(f(1)) == 2
    assert len(prompts) == 1 and "ok-wf" in prompts[0]

    monkeypatch.setattr("builtins.input", lambda msg: "n")
    try:
        with lzy.workflow("no-wf", interactive=True):
            int(f(1))
        raise AssertionError("expected WorkflowAbortedError")
    except WorkflowAbortedError:
        pass


def test_client_abort(lzy):
    """wf.abort() drops queued calls and surfaces WorkflowAbortedError
    (reference: AbortWorkflow/StopGraph)."""
    from lzy_amd.exceptions import WorkflowAbortedError

    ran = []

    @op
    def record(x: int) -> int:
        ran.append(x)
        return x

    try:
        with lzy.workflow("abort-wf", interactive=False) as wf:
            record(1)  # queued, never executed
            wf.abort("changed my mind")
        raise AssertionError("expected WorkflowAbortedError")
    except WorkflowAbortedError as e:
        assert "changed my mind" in str(e)
    assert ran == []


def test_api_v1_compat_surface(lzy):
    """Reference-shaped imports resolve (pylzy lzy.api.v1 parity)."""
    from lzy_amd.api.v1 import (  # noqa: F401
        File, GpuPoolRuntime, LocalRuntime, Lzy, LzyEnvironment,
        LzyWorkflow, Provisioning, Runtime, cpu, env_vars, gpu,
        lzy_auth, op as op_, whiteboard, whiteboard_,
    )

    @op_
    def f(x: int) -> int:
        return x * 3

    with Lzy(runtime=LocalRuntime()).workflow("compat", interactive=False):
        assert int(f(2)) == 6


def test_file_open_method(tmp_path):
    from lzy_amd import File

    f = File(tmp_path / "x.txt")
    with f.open("w") as fh:
        fh.write("via open")
    with f.open("r") as fh:
        assert fh.read() == "via open"


def test_env_variable_merge_precedence(tmp_path, monkeypatch):
    """Env merging (reference: LzyEnvironment.combine, test_op_params
    test_workflow_env/test_op_env): Lzy < workflow < op; applied around
    the op body and restored after."""
    import os as _os

    monkeypatch.setenv("LZY_AMD_STORAGE", str(tmp_path / "s"))
    from lzy_amd import Lzy, LzyEnvironment
    from lzy_amd.env.shortcuts import env_vars
    from lzy_amd.runtime.local import LocalRuntime

    seen = {}

    @op(env=env_vars(A="op", C="op"))
    def probe(x: int) -> int:
        seen.update({k: _os.environ.get(k) for k in ("A", "B", "C", "D")})
        return x

    lzy = Lzy(runtime=LocalRuntime()).with_env_variables(A="lzy", B="lzy", D="lzy")
    wf_env = env_vars(B="wf", C="wf")
    with lzy.workflow("envwf", env=wf_env, interactive=False):
        int(probe(1))

    assert seen["A"] == "op"    # op overrides all
    assert seen["B"] == "wf"    # workflow overrides Lzy
    assert seen["C"] == "op"    # op overrides workflow
    assert seen["D"] == "lzy"   # Lzy-level reaches the op
    assert _os.environ.get("A") != "op"  # restored afterwards


def test_container_of_proxies_as_arg(lzy):
    """A list of lazy results passed to another op materializes correctly
    (reference: materialize_if_sequence_of_proxies, proxy_adapter)."""

    @op
    def one(x: int) -> int:
        return x + 1

    @op
    def total(xs: list) -> int:
        return sum(xs)

    with lzy.workflow("seq-wf", interactive=False):
        parts = [one(i) for i in range(4)]
        s = total(parts)
        assert int(s) == 1 + 2 + 3 + 4


def test_concurrent_client_threads_serialize(tmp_path, monkeypatch):
    """Two client threads sharing one runtime run their workflows
    one-at-a-time (single-flight) with correct results."""
    import threading

    monkeypatch.setenv("LZY_AMD_STORAGE", str(tmp_path / "s"))
    from lzy_amd import Lzy
    from lzy_amd.runtime.local import LocalRuntime

    lzy = Lzy(runtime=LocalRuntime())
    results = {}

    @op
    def mul(a: int, b: int) -> int:
        return a * b

    def client(k: int):
        with lzy.workflow(f"thr-{k}", interactive=False):
            results[k] = int(mul(k, 10))

    ths = [threading.Thread(target=client, args=(k,)) for k in range(4)]
    for t in ths:
        t.start()
    for t in ths:
        t.join(timeout=60)
    assert results == {0: 0, 1: 10, 2: 20, 3: 30}


def test_runtime_survives_exit_time_failures(lzy):
    """Failures surfacing at the exit barrier (or whiteboard finalize)
    still release the runtime: the next workflow runs normally."""
    from lzy_amd.exceptions import LzyExecutionError

    @op
    def boom(x: int) -> int:
        raise RuntimeError("late boom")

    @op
    def fine(x: int) -> int:
        return x + 1

    # op failure that is never materialized in the body -> raises at exit
    try:
        with lzy.workflow("late-fail-wf", interactive=False):
            boom(1)  # lazy, untouched
        raise AssertionError("expected LzyExecutionError")
    except LzyExecutionError:
        pass

    # the SAME runtime must accept the next workflow (no leaked lock)
    with lzy.workflow("after-late-fail", interactive=False):
        assert int(fine(1)) == 2
