"""Crash-resume: a killed run resumes from persisted results
(reference restart-test semantics)."""
import os
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent


def _run(tmp_path, crash: bool):
    env = dict(os.environ)
    env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    env["MARK_DIR"] = str(tmp_path / "marks")
    env["PYTHONPATH"] = str(ROOT) + os.pathsep + env.get("PYTHONPATH", "")
    env["CRASH"] = "1" if crash else "0"
    os.makedirs(env["MARK_DIR"], exist_ok=True)
    return subprocess.run(
        [sys.executable, "tests/crash_script.py"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=120,
    )


def _count(tmp_path, name) -> int:
    p = tmp_path / "marks" / name
    return len(p.read_text()) if p.exists() else 0


def test_crash_then_resume(tmp_path):
    res1 = _run(tmp_path, crash=True)
    assert res1.returncode == 42, f"expected injected crash, rc={res1.returncode}"
    assert _count(tmp_path, "stage1") == 1
    assert _count(tmp_path, "stage2") == 1
    assert _count(tmp_path, "stage3") == 0  # never ran

    res2 = _run(tmp_path, crash=False)
    assert res2.returncode == 0, res2.stderr[-2000:]
    assert "CRASH-SCRIPT-DONE" in res2.stdout
    # stages 1-2 served from the persisted cache, only stage3 executed
    assert _count(tmp_path, "stage1") == 1
    assert _count(tmp_path, "stage2") == 1
    assert _count(tmp_path, "stage3") == 1


def test_journal_written(tmp_path):
    import json

    env = dict(os.environ)
    env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    env["MARK_DIR"] = str(tmp_path / "marks")
    env["PYTHONPATH"] = str(ROOT) + os.pathsep + env.get("PYTHONPATH", "")
    env["CRASH"] = "0"
    os.makedirs(env["MARK_DIR"], exist_ok=True)
    res = subprocess.run(
        [sys.executable, "tests/crash_script.py"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=120,
    )
    assert res.returncode == 0
    import glob
    import tempfile

    journals = sorted(
        glob.glob(os.path.join(tempfile.gettempdir(), "lzy_amd_journal", "crashy-*")),
        key=os.path.getmtime,
    )
    assert journals, "journal file missing"
    from lzy_amd.sched import Journal

    states = Journal.replay(journals[-1])
    assert len(states) == 3
    assert all(s == "done" for s in states.values())


def test_pool_crash_then_resume(tmp_path):
    """Crash the whole pool mid-graph; a fresh pool resumes from the
    result cache (GpuPoolRuntime durable-resume path)."""
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    marks = tmp_path / "marks"
    marks.mkdir()

    def run(crash: bool):
        import socket

        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
        env = dict(os.environ)
        env.update(
            LZY_AMD_STORAGE=str(tmp_path / "storage"),
            PYTHONPATH=str(root) + os.pathsep + env.get("PYTHONPATH", ""),
            MARK_DIR=str(marks),
            CRASH="1" if crash else "0",
        )
        env.pop("RANK", None)
        env.pop("WORLD_SIZE", None)
        return subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node=2", "--master-addr", "127.0.0.1",
             "--master-port", str(port), "tests/pool_crash_script.py"],
            cwd=root, env=env, capture_output=True, text=True, timeout=180,
        )

    r1 = run(crash=True)
    assert r1.returncode != 0  # the hard exit propagates
    def count(stage):
        return sum(1 for f in marks.iterdir() if f.name.startswith(stage))
    assert count("stage1") == 1 and count("stage2") == 1 and count("stage3") == 0

    r2 = run(crash=False)
    assert r2.returncode == 0, r2.stdout[-2000:] + r2.stderr[-2000:]
    assert "RESULT=15" in r2.stdout and "POOL-RESUME-OK" in r2.stdout
    # stage1/2 served from cache: no second execution
    assert count("stage1") == 1 and count("stage2") == 1 and count("stage3") == 1


def test_repeated_crash_resume_chain(tmp_path):
    """Reference stress-runner semantics: kill, restart, kill again,
    restart, complete — each restart resumes from the persisted results
    and the crash point walks forward one stage per run."""
    env = dict(os.environ)
    env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    env["MARK_DIR"] = str(tmp_path / "marks")
    env["PYTHONPATH"] = str(ROOT) + os.pathsep + env.get("PYTHONPATH", "")
    os.makedirs(env["MARK_DIR"], exist_ok=True)

    def run(crash_at):
        e = dict(env)
        if crash_at is not None:
            e["CRASH_AT"] = str(crash_at)
        return subprocess.run(
            [sys.executable, "tests/crash_script.py"],
            cwd=ROOT, env=e, capture_output=True, text=True, timeout=120,
        )

    # run 1: dies right after the first executed op (stage1)
    assert run(0).returncode == 42
    assert _count(tmp_path, "stage1") == 1
    assert _count(tmp_path, "stage2") == 0
    # run 2: stage1 cached; dies after the next executed op (stage2)
    assert run(0).returncode == 42
    assert _count(tmp_path, "stage1") == 1  # not re-executed
    assert _count(tmp_path, "stage2") == 1
    assert _count(tmp_path, "stage3") == 0
    # run 3: completes; only stage3 executes
    r = run(None)
    assert r.returncode == 0 and "CRASH-SCRIPT-DONE" in r.stdout
    assert (_count(tmp_path, "stage1"), _count(tmp_path, "stage2"),
            _count(tmp_path, "stage3")) == (1, 1, 1)
