"""End-to-end scenario tier (reference: test/.../PyApiTest.java +
pylzy/tests/scenarios/<name>/ — each scenario is a directory with a
script and an ``expected_stdout`` asserted line-by-line,
PythonContextTests.java:197-246).

Here each scenario under tests/scenarios/<name>/ has ``__init__.py``
(the script, run as ``python -m tests.scenarios.<name>`` in a fresh
process with an isolated storage root) and ``expected_stdout``.
Expected lines match exactly, except lines starting with ``~`` which are
regex-matched, and ``...`` which skips any number of output lines until
the next expected line matches.
"""
import os
import re
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parent.parent
SCEN = Path(__file__).resolve().parent / "scenarios"

_names = sorted(
    d.name for d in SCEN.iterdir()
    if d.is_dir() and (d / "expected_stdout").exists()
) if SCEN.exists() else []


def _match(expected_lines, actual_lines, scenario):
    ei, ai = 0, 0
    while ei < len(expected_lines):
        exp = expected_lines[ei]
        if exp == "...":
            ei += 1
            if ei >= len(expected_lines):
                return  # trailing ... matches the rest
            nxt = expected_lines[ei]
            while ai < len(actual_lines) and not _line_ok(nxt, actual_lines[ai]):
                ai += 1
            if ai >= len(actual_lines):
                pytest.fail(
                    f"[{scenario}] expected line {ei} not found after '...': "
                    f"{nxt!r}\n--- actual ---\n" + "\n".join(actual_lines)
                )
            continue
        if ai >= len(actual_lines) or not _line_ok(exp, actual_lines[ai]):
            got = actual_lines[ai] if ai < len(actual_lines) else "<eof>"
            pytest.fail(
                f"[{scenario}] line {ai}: expected {exp!r}, got {got!r}\n"
                "--- actual ---\n" + "\n".join(actual_lines)
            )
        ei += 1
        ai += 1


def _line_ok(exp: str, actual: str) -> bool:
    if exp.startswith("~"):
        return re.fullmatch(exp[1:], actual) is not None
    return exp == actual


@pytest.mark.parametrize("name", _names)
def test_scenario(name, tmp_path):
    env = dict(os.environ)
    env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    env["PYTHONPATH"] = str(ROOT) + os.pathsep + env.get("PYTHONPATH", "")
    env["LZY_SCENARIO_TMP"] = str(tmp_path)
    r = subprocess.run(
        [sys.executable, str(SCEN / name / "__init__.py")],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, (
        f"[{name}] rc={r.returncode}\n--- stdout ---\n{r.stdout[-4000:]}"
        f"\n--- stderr ---\n{r.stderr[-4000:]}"
    )
    expected = (SCEN / name / "expected_stdout").read_text().splitlines()
    # op std-logs reach the client prefixed "[LZY-<task>] " (reference:
    # "[LZY-REMOTE-...]", runtime.py:283-301) — strip the prefix so
    # expectations read like plain program output
    actual = [re.sub(r"^\[LZY-[^\]]*\] ", "", l) for l in r.stdout.splitlines()]
    _match(expected, actual, name)


# the same scenario scripts under the pool engine (torchrun world 2):
# "one script, both runtimes" — GpuPoolRuntime is picked from the launch
# context, ops run on worker ranks, values cross the gloo/RCCL data plane
_POOL_SCENARIOS = [
    "complex_graph", "repeated_ops_use_cache", "fully_cached_graph",
    "exec_fail", "cached_exception", "custom_serializer", "file_test",
    "nested_workflows", "whiteboards", "two_execution_one_wf",
    "exception_serialize", "subprocess_with_startup", "stream_merge",
]


@pytest.mark.parametrize("name", [n for n in _POOL_SCENARIOS if n in _names])
def test_scenario_pool(name, tmp_path):
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    env = dict(os.environ)
    env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    env["PYTHONPATH"] = str(ROOT) + os.pathsep + env.get("PYTHONPATH", "")
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    log_dir = tmp_path / "ranklogs"
    log_dir.mkdir()
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         # per-rank output files: cross-rank interleaving would split
         # lines mid-write on a shared console
         "--log-dir", str(log_dir), "--redirects", "3",
         str(SCEN / name / "__init__.py")],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=300,
    )
    rank0_out = ""
    for f in sorted(log_dir.rglob("stdout*")):
        if f.parent.name == "0" or "/0/" in str(f):
            rank0_out = f.read_text()
            break
    assert r.returncode == 0, (
        f"[{name}/pool] rc={r.returncode}\n--- rank0 stdout ---\n"
        f"{rank0_out[-4000:]}\n--- stderr ---\n{r.stderr[-4000:]}"
    )
    expected = (SCEN / name / "expected_stdout").read_text().splitlines()
    actual = [
        re.sub(r"^\[LZY-[^\]]*\] ", "", l)
        for l in rank0_out.splitlines()
        # gloo prints its connection banner on stdout at pg init
        if l.strip()
        and not l.startswith("[Gloo]") and " is connected to " not in l
        and not re.match(r"^[WIE]\d{4}", l)
    ]
    _match(expected, actual, f"{name}/pool")
