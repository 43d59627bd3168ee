"""An op that spawns a subprocess which itself uses lzy (reference
scenario: subprocess_with_startup — the DDP-subprocess guard)."""
import os
import subprocess
import sys

from lzy_amd import Lzy, op

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(os.path.dirname(os.path.dirname(HERE)))


@op
def run_child(x: int) -> int:
    env = {**os.environ, "LZY_REPO_ROOT": ROOT,
           "PYTHONPATH": ROOT + os.pathsep + os.environ.get("PYTHONPATH", "")}
    r = subprocess.run(
        [sys.executable, os.path.join(HERE, "child.py")],
        env=env, capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stderr[-1500:]
    sys.stdout.write(r.stdout)
    return x


if __name__ == "__main__":
    with Lzy().workflow("parent-wf", interactive=False):
        print(f"parent={int(run_child(1))}", flush=True)
