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
