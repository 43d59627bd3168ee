"""Multi-process pool runtime tests: world_size 2 over gloo on CPU
(the distributed-correctness harness the reference runs as in-thread
multi-service contexts — SURVEY.md §4.3; here: real processes, real
torch.distributed, no GPU required)."""
import os
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parent.parent


def _run_distributed(script: str, nproc: int, tmp_path, extra_env=None, timeout=180):
    env = dict(os.environ)
    env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    env["PYTHONPATH"] = str(ROOT) + os.pathsep + env.get("PYTHONPATH", "")
    env["MASTER_ADDR"] = "127.0.0.1"
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    if extra_env:
        env.update(extra_env)
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={nproc}",
        "--master-addr", "127.0.0.1", "--master-port", str(port),
        script,
    ]
    return subprocess.run(
        cmd, cwd=ROOT, env=env, capture_output=True, text=True, timeout=timeout
    )


def test_pool_two_ranks(tmp_path):
    res = _run_distributed("tests/pool_script.py", 2, tmp_path)
    if res.returncode != 0:
        print("STDOUT:", res.stdout[-4000:])
        print("STDERR:", res.stderr[-4000:])
    assert res.returncode == 0
    assert "POOL-SCRIPT-OK" in res.stdout


def test_pool_two_ranks_coalesced_works(tmp_path):
    """Same battery under a simulated NCCL-coalesced batch_isend_irecv
    (ONE work per group, the shape RCCL returns on the real node):
    Transport.issue must normalize so settle slicing stays correct."""
    res = _run_distributed("tests/pool_script_coalesce.py", 2, tmp_path)
    if res.returncode != 0:
        print("STDOUT:", res.stdout[-4000:])
        print("STDERR:", res.stderr[-4000:])
    assert res.returncode == 0
    assert "POOL-SCRIPT-OK" in res.stdout


def test_pool_single_rank_inprocess(tmp_path, monkeypatch):
    """ws=1 degenerates to driver-only: full pipeline without dist."""
    env = dict(os.environ)
    env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    env["PYTHONPATH"] = str(ROOT) + os.pathsep + env.get("PYTHONPATH", "")
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    res = subprocess.run(
        [sys.executable, "tests/pool_script_single.py"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=120,
    )
    if res.returncode != 0:
        print("STDOUT:", res.stdout[-4000:])
        print("STDERR:", res.stderr[-4000:])
    assert res.returncode == 0
    assert "SINGLE-OK" in res.stdout


def _run_death_scenario(tmp_path, extra_env):
    # launch ranks manually: torchrun would kill the driver the moment
    # rank 1 exits, hiding exactly the recovery we want to observe
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    base_env = dict(os.environ)
    base_env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    base_env["PYTHONPATH"] = str(ROOT) + os.pathsep + base_env.get("PYTHONPATH", "")
    base_env.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), WORLD_SIZE="2"
    )
    base_env.update(extra_env)
    procs = []
    for rank in (0, 1):
        env = dict(base_env)
        env["RANK"] = str(rank)
        env["LOCAL_RANK"] = str(rank)
        procs.append(subprocess.Popen(
            [sys.executable, "tests/pool_script_death.py"],
            cwd=ROOT, env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            text=True,
        ))
    out0, err0 = procs[0].communicate(timeout=120)
    procs[1].wait(timeout=30)
    assert procs[1].returncode == 7  # the injected hard exit
    return out0, err0


def test_worker_death_recovers(tmp_path):
    """Default: inflight tasks of the dead rank re-dispatch to survivors
    and the workflow COMPLETES (reference: SlotsService failover)."""
    out0, err0 = _run_death_scenario(tmp_path, {})
    assert "DEATH-RECOVERED" in out0, out0[-3000:] + err0[-2000:]


def test_worker_death_detected_without_retries(tmp_path):
    """task_retries=0: recovery off — the failure must surface, not hang."""
    out0, err0 = _run_death_scenario(tmp_path, {"LZY_TASK_RETRIES": "0"})
    assert "DEATH-DETECTED" in out0, out0[-3000:] + err0[-2000:]
    assert "DEATH-RECOVERED" not in out0


def test_streamed_merge_tree_world4(tmp_path):
    """pair_reduce trees fold into one chunk-streamed plan; results match
    the op-by-op reference; interior outputs stay readable; mismatched
    components break apart gracefully."""
    r = _run_distributed("tests/pool_script_streammerge.py", 4, tmp_path,
                         timeout=300)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "STREAMMERGE-OK" in r.stdout


def test_transfer_source_death_durable_failover(tmp_path):
    """A transfer's SOURCE rank dies mid-send; the consumer's settle
    times out and the driver re-sources the cached input from the
    durable tier, re-dispatching instead of failing (reference:
    transferFailed -> storage peer, SlotsService.java:191-240)."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    base_env = dict(os.environ)
    base_env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    base_env["PYTHONPATH"] = str(ROOT) + os.pathsep + base_env.get("PYTHONPATH", "")
    base_env.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), WORLD_SIZE="3"
    )
    procs = []
    for rank in range(3):
        env = dict(base_env)
        env["RANK"] = str(rank)
        env["LOCAL_RANK"] = str(rank)
        procs.append(subprocess.Popen(
            [sys.executable, "tests/pool_script_xferdeath.py"],
            cwd=ROOT, env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            text=True,
        ))
    out0, err0 = procs[0].communicate(timeout=180)
    for p in procs[1:]:
        try:
            p.wait(timeout=60)
        except subprocess.TimeoutExpired:
            p.kill()
    assert "XFERDEATH-RECOVERED" in out0, out0[-3000:] + err0[-2000:]
    # the injected death must actually have fired (rank 1 exits 9)
    assert procs[1].returncode == 9, procs[1].returncode


def test_worker_death_during_stream_plan(tmp_path):
    """A participant dying mid-plan fails the folded tasks promptly and
    raises on the client — no hang, survivors' stale events dropped."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    base_env = dict(os.environ)
    base_env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    base_env["PYTHONPATH"] = str(ROOT) + os.pathsep + base_env.get("PYTHONPATH", "")
    base_env.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), WORLD_SIZE="4"
    )
    procs = []
    for rank in range(4):
        env = dict(base_env)
        env["RANK"] = str(rank)
        env["LOCAL_RANK"] = str(rank)
        procs.append(subprocess.Popen(
            [sys.executable, "tests/pool_script_plandeath.py"],
            cwd=ROOT, env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            text=True,
        ))
    out0, err0 = procs[0].communicate(timeout=150)
    for p in procs[1:]:
        try:
            p.wait(timeout=60)
        except subprocess.TimeoutExpired:
            p.kill()
    assert "PLANDEATH-DETECTED" in out0, out0[-3000:] + err0[-2000:]
    assert "PLANDEATH-NOT-DETECTED" not in out0


def test_murder_fuzz_pool_survives(tmp_path):
    """A rank hard-exits mid-campaign; affected seeds fail typed or
    retry, and every later seed completes on the survivors."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    base_env = dict(os.environ)
    base_env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    base_env["PYTHONPATH"] = str(ROOT) + os.pathsep + base_env.get("PYTHONPATH", "")
    base_env.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), WORLD_SIZE="3",
        FUZZ_MURDER="1:6", FUZZ_ROUNDS="25", FUZZ_BASE_SEED="52000",
    )
    procs = []
    for rank in range(3):
        env = dict(base_env)
        env["RANK"] = str(rank)
        env["LOCAL_RANK"] = str(rank)
        procs.append(subprocess.Popen(
            [sys.executable, "tests/pool_script_fuzz.py"],
            cwd=ROOT, env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            text=True,
        ))
    out0, err0 = procs[0].communicate(timeout=240)
    for p in procs[1:]:
        try:
            p.wait(timeout=60)
        except subprocess.TimeoutExpired:
            p.kill()
    assert "FUZZ-OK" in out0, out0[-3000:] + err0[-2000:]
    assert procs[1].returncode == 66  # the injected murder fired


def test_live_remote_log_tail(tmp_path):
    """A remote rank's print appears on the driver console BEFORE the op
    finishes (live ReadStdSlots-style streaming, not completion-time)."""
    r = _run_distributed("tests/pool_script_livelogs.py", 2, tmp_path)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "LIVELOGS-OK" in r.stdout


def test_concurrent_gangs_world8(tmp_path):
    """Two+ disjoint gpu_count=2 gangs and singles run simultaneously
    (driver-sequenced group creation, disjoint-rank admission)."""
    r = _run_distributed("tests/pool_script_gangs.py", 8, tmp_path, timeout=300)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "GANGS-OK" in r.stdout


def test_sigint_mid_barrier(tmp_path):
    """Ctrl-C in the blocked barrier stops the graph promptly and the
    interrupt propagates (queued ops cancelled, not drained)."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    base_env = dict(os.environ)
    base_env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    base_env["PYTHONPATH"] = str(ROOT) + os.pathsep + base_env.get("PYTHONPATH", "")
    base_env.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), WORLD_SIZE="2"
    )
    procs = []
    for rank in (0, 1):
        env = dict(base_env)
        env["RANK"] = str(rank)
        env["LOCAL_RANK"] = str(rank)
        procs.append(subprocess.Popen(
            [sys.executable, "tests/pool_script_sigint.py"],
            cwd=ROOT, env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            text=True,
        ))
    out0, err0 = procs[0].communicate(timeout=120)
    for p in procs[1:]:
        try:
            p.wait(timeout=30)
        except subprocess.TimeoutExpired:
            p.kill()
    assert "SIGINT-HANDLED" in out0, out0[-2000:] + err0[-2000:]


def test_stop_graph_mid_flight(tmp_path):
    """Abort of a slow fan-out mid-barrier: queued tasks cancelled on all
    ranks, barrier raises promptly, pool stays usable (StopGraph)."""
    r = _run_distributed("tests/pool_script_stop.py", 2, tmp_path)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "STOP-OK" in r.stdout


def test_chunked_transport_offset_resume(tmp_path):
    r = _run_distributed("tests/chunked_transport_script.py", 2, tmp_path)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "CHUNKED-TRANSPORT-OK" in r.stdout


def test_bench_dag_four_ranks(tmp_path):
    """The driver's 8-GPU scale shape, at world 4 on gloo: fan-out,
    cross-rank tree merge, scalar gather, repeated steps."""
    r = _run_distributed("tests/bench_dag_script.py", 4, tmp_path, timeout=240)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "BENCH-DAG-OK" in r.stdout


def test_pool_world4_gangs_and_chunks(tmp_path):
    r = _run_distributed("tests/pool_script_world4.py", 4, tmp_path, timeout=240)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "WORLD4-OK" in r.stdout


def test_bench_dag_eight_ranks(tmp_path):
    """Exactly the driver's 8-GPU SCALE shape, world 8 on gloo."""
    r = _run_distributed(
        "tests/bench_dag_script.py", 8, tmp_path,
        extra_env={"LZY_BENCH_SHARD_MB": "1"}, timeout=300,
    )
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "BENCH-DAG-OK" in r.stdout


def test_chain_dispatch(tmp_path):
    r = _run_distributed("tests/pool_script_chain.py", 2, tmp_path, timeout=240)
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "CHAIN-OK" in r.stdout


@pytest.mark.parametrize("world", [2, 3])
def test_pool_fuzz_random_dags(world, tmp_path):
    """Seeded random DAGs through the pool; every sink checked against a
    pure-python evaluation (scheduler/chaining/transfer fuzz)."""
    r = _run_distributed(
        "tests/pool_script_fuzz.py", world, tmp_path,
        extra_env={"FUZZ_BASE_SEED": str(2000 + world), "FUZZ_ROUNDS": "10"},
        timeout=300,
    )
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "FUZZ-OK" in r.stdout


def test_pool_fuzz_chaos_abort(tmp_path):
    """Random mid-flight aborts over the DAG fuzz: every run either
    completes with correct sinks or raises WorkflowAbortedError, and the
    pool stays usable across seeds (StopGraph under fuzz)."""
    r = _run_distributed(
        "tests/pool_script_fuzz.py", 2, tmp_path,
        extra_env={"FUZZ_BASE_SEED": "4100", "FUZZ_ROUNDS": "12",
                   "FUZZ_CHAOS_ABORT": "1"},
        timeout=300,
    )
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    assert "FUZZ-OK" in r.stdout
