"""The driver parses ONE JSON line from bench.py — pin the contract
(field names, types, aggregate semantics) on the CPU path so a refactor
cannot silently break the round-end harness."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract(tmp_path):
    env = dict(os.environ)
    env["LZY_BENCH_SHARD_MB"] = "2"
    env["LZY_AMD_STORAGE"] = str(tmp_path / "storage")
    env["PYTHONPATH"] = ROOT + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"),
         "--steps", "2", "--warmup", "1"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    line = r.stdout.strip().splitlines()[-1]
    d = json.loads(line)

    assert d["metric"] == "workflow_makespan_s_8stage_dag"
    assert isinstance(d["value"], float) and d["value"] > 0
    assert d["unit"] == "s"
    assert d["n_gpus"] == 1
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is False
    assert d["scaling"] == "weak"
    assert d["vs_baseline"] is None  # reference publishes no number
    assert d["data"] == "synthetic"
    assert isinstance(d["ms_per_step"], float)
    cfg = d["config"]
    # steps * ms_per_step reproduces the timed region (driver's check)
    assert abs(d["steps"] * d["ms_per_step"] / 1000.0
               - cfg["timed_region_s"]) < 0.05 * cfg["timed_region_s"] + 1e-6
    # value is the per-DAG makespan of repeats-declared work
    assert cfg["dag_repeats_per_step"] >= 1
    assert abs(cfg["dag_ms"] - d["value"] * 1000.0) < 1e-6
    for key in ("model", "global_batch", "seq_len", "parallelism",
                "shard_mb", "ops_per_dag", "transfers", "stage_ms"):
        assert key in cfg, key
    assert set(cfg["stage_ms"]) >= {"ingest", "preprocess", "augment",
                                    "train_step", "checksum", "evaluate",
                                    "report"}
