"""The op result cache survives across separate process executions
(reference scenario: repeated_execs_use_cache)."""
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
ROOT = os.path.dirname(os.path.dirname(os.path.dirname(HERE)))

if __name__ == "__main__":
    env = {**os.environ, "LZY_REPO_ROOT": ROOT,
           "PYTHONPATH": ROOT + os.pathsep + os.environ.get("PYTHONPATH", "")}
    for _ in range(2):
        r = subprocess.run(
            [sys.executable, os.path.join(HERE, "exec_once.py")],
            env=env, capture_output=True, text=True, timeout=120,
        )
        assert r.returncode == 0, r.stderr[-1500:]
        sys.stdout.write(r.stdout)
