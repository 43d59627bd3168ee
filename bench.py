"""Flagship benchmark — driver contract (BASELINE.json).

Metric: workflow makespan (s) + per-op scheduling overhead for an 8-stage
DAG, weak-scaled across 1/2/4/8 MI355X (fan-out width = N GPUs, per-GPU
work fixed).

One *step* = one full execution of the 8-stage workflow DAG:

  1. ingest     (xN)  synthetic shard tensor in HBM (HIP fill kernel)
  2. preprocess (xN)  normalize (memory-bound, rank-local)
  3. augment    (xN)  scale+shift (memory-bound, rank-local)
  4. train      (xN)  fwd+bwd+optimizer step of an MLP on a shard slice
  5. checksum   (xN)  device-side content hash (HIP checksum kernel)
  6. merge      (tree) pairwise shard reduction  -> xGMI/RCCL transfers
  7. evaluate   (x1)  loss metric on merged shard
  8. report     (x1)  scalar summary gather

The reference's structural floor for the same shape is >=1 s dispatch
tick + 10 s completion poll + S3 round trips per edge; this
runtime dispatches in-process and moves tensors over xGMI.

Usage (driver launches):  python bench.py --gpus N --steps K --warmup W
For N>1 it runs under `python -m torch.distributed.run --nproc-per-node N
--master-addr 127.0.0.1 ...` — one rank per GPU over RCCL.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

from lzy_amd import Lzy, op
from lzy_amd.utils.metrics import METRICS

# ---------------------------------------------------------------------------
# op definitions (module level: cloudpickled to workers)
# ---------------------------------------------------------------------------

SHARD_MB = int(os.environ.get("LZY_BENCH_SHARD_MB", "0"))  # 0 -> auto
TRAIN_BATCH = int(os.environ.get("LZY_BENCH_TRAIN_BATCH", "8192"))
TRAIN_DIM = int(os.environ.get("LZY_BENCH_TRAIN_DIM", "4096"))


def _device() -> torch.device:
    if torch.cuda.is_available():
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def _shard_elems() -> int:
    mb = SHARD_MB or (1024 if torch.cuda.is_available() else 8)
    return (mb << 20) // 2  # bf16


@op
def ingest(shard_idx: int, seed: int) -> torch.Tensor:
    dev = _device()
    n = _shard_elems()
    t = torch.empty(n, dtype=torch.bfloat16, device=dev)
    if dev.type == "cuda":
        from lzy_amd.ops import fill_pattern

        # mask16=0x3FFF clamps bf16 exponents in the same pass: finite
        # positive synthetic data, one HBM write instead of three passes
        fill_pattern(t, seed=seed * 1000 + shard_idx, mask16=0x3FFF)
    else:
        t.normal_()
    return t


@op
def preprocess(t: torch.Tensor) -> torch.Tensor:
    if t.is_cuda:
        # fused two-pass normalize (HIP): 3.2 GB of HBM traffic per GiB
        # shard vs ~14 GB for the unfused torch chain
        from lzy_amd.ops import normalize

        return normalize(t)
    x = t.float()
    x = (x - x.mean()) / (x.std() + 1e-6)
    return x.to(torch.bfloat16)


@op
def augment(t: torch.Tensor) -> torch.Tensor:
    if t.is_cuda:
        from lzy_amd.ops import scale_shift

        return scale_shift(t, 1.0009765625, 0.125)
    return t * 1.0009765625 + 0.125


_MODEL_CACHE = {}


def _get_model(d: int, dev: torch.device, dtype: torch.dtype):
    # persistent training state per worker (a real training loop keeps its
    # model resident in HBM; re-initializing 134 MB of weights per step
    # would be benchmarking weight init, not the runtime)
    key = (d, dev.type, dtype)
    hit = _MODEL_CACHE.get(key)
    if hit is None:
        model = torch.nn.Sequential(
            torch.nn.Linear(d, d, device=dev, dtype=dtype),
            torch.nn.GELU(),
            torch.nn.Linear(d, d, device=dev, dtype=dtype),
        )
        opt = torch.optim.SGD(model.parameters(), lr=1e-3)
        hit = (model, opt)
        _MODEL_CACHE[key] = hit
    return hit


@op
def train_step(t: torch.Tensor) -> float:
    dev = _device()
    d = TRAIN_DIM if dev.type == "cuda" else 256
    b = TRAIN_BATCH if dev.type == "cuda" else 512
    # debug/test harnesses shrink the shard (LZY_BENCH_SHARD_MB): clamp
    # the batch to what the shard holds.  The driver's real runs use the
    # default 1 GiB shard, which always fills the full batch.
    b = min(b, max(1, t.numel() // d))
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
    model, opt = _get_model(d, dev, dtype)
    x = t[: b * d].reshape(b, d).to(dtype)

    def step_fn(xin: torch.Tensor) -> torch.Tensor:
        opt.zero_grad(set_to_none=True)
        loss = model(xin).float().square().mean()
        loss.backward()
        opt.step()
        return loss.detach()

    # fwd+bwd+optimizer is ~60 small launches: captured once as a
    # hipGraph and replayed as ONE launch per step (utils/hipgraph.py)
    global _TRAIN_GRAPH
    try:
        sg = _TRAIN_GRAPH
    except NameError:
        from lzy_amd.utils.hipgraph import StepGraph

        sg = _TRAIN_GRAPH = StepGraph(step_fn)
    return float(sg.run(x).item())


@op
def checksum(t: torch.Tensor) -> int:
    if t.is_cuda:
        from lzy_amd.ops import device_checksum

        return device_checksum(t)
    from lzy_amd.snapshot import _hash_bytes

    return int(_hash_bytes(t.float().numpy().tobytes()[:1 << 20]), 16)


@op(pair_reduce=(0.5, 0.5))
def merge(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    # pair_reduce declares the semantics (0.5*a + 0.5*b), letting the
    # pool fold the whole merge tree into ONE chunk-streamed reduction
    # plan over xGMI (channels/treeplan.py): a chunk combined at level k
    # is forwarded to level k+1 immediately, so tree latency collapses
    # from log2(N) transfers to ~one.  This body remains the semantic
    # reference (LocalRuntime / same-rank / fallback paths).
    if a.is_cuda and b.is_cuda:
        from lzy_amd.ops import axpby

        return axpby(a, b, 0.5, 0.5)
    return (a.float() + b.float()).mul_(0.5).to(torch.bfloat16)


@op
def evaluate(t: torch.Tensor) -> float:
    if t.is_cuda:
        from lzy_amd.ops import abs_mean

        return abs_mean(t)
    return float(t.float().abs().mean().item())


@op
def report(loss_sum: float, eval_score: float, checks: int) -> float:
    return loss_sum + eval_score + (checks % 97)


# ---------------------------------------------------------------------------
# the 8-stage DAG
# ---------------------------------------------------------------------------

def run_dag(lzy: Lzy, width: int, step_idx: int) -> float:
    with lzy.workflow(f"bench-{step_idx}") as wf:
        shards = [ingest(i, step_idx) for i in range(width)]
        pre = [preprocess(s) for s in shards]
        aug = [augment(p) for p in pre]
        losses = [train_step(a) for a in aug]
        checks = [checksum(a) for a in aug]
        # pairwise tree reduction (cross-rank xGMI transfers)
        layer = aug
        while len(layer) > 1:
            nxt = []
            for i in range(0, len(layer) - 1, 2):
                nxt.append(merge(layer[i], layer[i + 1]))
            if len(layer) % 2:
                nxt.append(layer[-1])
            layer = nxt
        ev = evaluate(layer[0])
        summary = report(
            sum(float(l) for l in losses),
            float(ev),
            sum(int(c) for c in checks) & 0x7FFFFFFF,
        )
        out = float(summary)
    return out


def _preflight(pool, n_gpus: int) -> None:
    """Fail fast with a rank-tagged diagnostic instead of a silent hang:
    every rank exercises every communicator (default, pg_data, pg_stream;
    CPU and CUDA) once, so a broken RCCL rendezvous surfaces here, not
    mid-benchmark.  Workers run it via the pool's preflight command."""
    try:
        pool.preflight()
    except BaseException as e:
        print(
            f"[bench rank {pool.rank}] PREFLIGHT FAILED "
            f"({type(e).__name__}: {e}) world={pool.world} "
            f"device={pool.device} "
            f"cuda_p2p={pool.agent.transport._cuda_p2p}",
            file=sys.stderr, flush=True,
        )
        raise


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=12)
    parser.add_argument("--warmup", type=int, default=3)
    args = parser.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    n_gpus = max(args.gpus, world)
    width = n_gpus  # weak scaling: one shard per GPU

    os.environ.setdefault(
        "LZY_AMD_STORAGE", os.path.join("/tmp", "lzy_amd_bench_storage")
    )

    from lzy_amd.runtime.pool import GpuPool, GpuPoolRuntime

    pool = GpuPool.get()  # workers serve here and never return
    _preflight(pool, n_gpus)
    runtime = GpuPoolRuntime()
    lzy = Lzy(runtime=runtime)

    dag_times = []
    for i in range(args.warmup):
        t_w0 = time.perf_counter()
        run_dag(lzy, width, i)
        dag_times.append(time.perf_counter() - t_w0)
    # steady-state estimate: the FASTEST warmup DAG (the first pays
    # one-time costs — kernel loads, hipGraph capture, group creation)
    t_dag_est = min(dag_times) if dag_times else 0.1

    # the timed region must be long enough for utilization samplers and
    # rocprof attribution to register (>= ~2 s): each *step* executes the
    # DAG `repeats` times so even the default 12 steps span >= ~2 s.
    # The reported metric stays the PER-DAG makespan; repeats is declared
    # in config (honest accounting, same work per DAG).
    repeats = max(1, min(64, int(0.18 / max(1e-4, t_dag_est))))

    def step(i: int) -> None:
        for r in range(repeats):
            run_dag(lzy, width, args.warmup + i * repeats + r)

    try:
        METRICS.reset()
        ts_a = pool.sync_all()
        t0 = time.perf_counter()
        for i in range(args.steps):
            step(i)
        ts_b = pool.sync_all()
        t1 = time.perf_counter()
    except BaseException as e:
        import traceback

        print(
            f"[bench rank {pool.rank}] STEP FAILED: {type(e).__name__}: {e}\n"
            + traceback.format_exc(),
            file=sys.stderr, flush=True,
        )
        raise

    # per-rank elapsed between the two barriers, max over ranks
    elapsed_by_rank = {r: ts_b[r] - ts_a[r] for r in ts_b if r in ts_a}
    elapsed = max(max(elapsed_by_rank.values(), default=t1 - t0), t1 - t0)

    n_dags = args.steps * repeats
    makespan_s = elapsed / n_dags          # the metric: per-DAG makespan
    dispatch = METRICS.timing_stats("lzy_dispatch")
    sched_overhead_ms = 1000.0 * dispatch.get("mean", 0.0)

    shard_mb = (SHARD_MB or (1024 if torch.cuda.is_available() else 8))
    n_ops = 5 * width + (width - 1) + 2

    transfers = int(METRICS.counter_value("lzy_transfers")
                    + METRICS.counter_value("lzy_transfers_ipc"))
    if width > 1 and transfers == 0:
        raise RuntimeError(
            f"N={width} but zero cross-rank transfers were recorded — "
            "the data plane did not engage; refusing to report a number "
            "that does not measure it"
        )

    # per-stage timing breakdown (rocprof/gpu_busy attribution aid)
    stage_ms = {}
    for name in sorted(getattr(METRICS, "_timings", {})):
        if name.startswith("lzy_op::"):
            st = METRICS.timing_stats(name)
            stage_ms[name[len("lzy_op::"):]] = round(st["mean"] * 1e3, 3)

    # effective per-edge bandwidth estimate from the streamed merge plans
    # (plan time ~= one pipelined shard transfer)
    plan_stats = METRICS.timing_stats("lzy_stream_plan_s")
    xgmi_gbps_est = None
    if plan_stats.get("count", 0) and plan_stats.get("mean", 0.0) > 0:
        xgmi_gbps_est = round(
            (shard_mb * (1 << 20)) / plan_stats["mean"] / 1e9, 2
        )

    if os.environ.get("LZY_BENCH_PROFILE"):
        import sys as _sys

        for name, xs in sorted(METRICS._timings.items()):
            if name.startswith(("lzy_op::", "lzy_wf_")) or name in ("lzy_dispatch", "lzy_graph_build", "lzy_task_overhead"):
                st = METRICS.timing_stats(name)
                print(
                    f"# {name}: n={st['count']} mean={st['mean']*1e3:.2f}ms "
                    f"p99={st['p99']*1e3:.2f}ms total={st['total']*1e3:.1f}ms",
                    file=_sys.stderr,
                )

    result = {
        "metric": "workflow_makespan_s_8stage_dag",
        "value": round(makespan_s, 6),
        "unit": "s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        # ms per driver step (repeats DAGs each) — steps * ms_per_step
        # reproduces the timed-region wall clock for the driver's check
        "ms_per_step": round(1000.0 * elapsed / args.steps, 3),
        "higher_is_better": False,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "sched_overhead_ms_per_op": round(sched_overhead_ms, 4),
        "config": {
            "model": "8-stage DAG (ingest/preprocess/augment/train/checksum/merge-tree/evaluate/report)",
            "global_batch": TRAIN_BATCH * width,
            "seq_len": TRAIN_DIM,
            "parallelism": f"dag-fanout{width}",
            "shard_mb": shard_mb,
            "ops_per_dag": n_ops,
            # one driver step = this many full DAG executions, so the
            # timed region spans >= ~2 s for utilization sampling; the
            # reported value is still the per-DAG makespan
            "dag_repeats_per_step": repeats,
            "dag_ms": round(1000.0 * makespan_s, 3),
            "timed_region_s": round(elapsed, 3),
            "dispatch_p99_ms": round(1000.0 * dispatch.get("p99", 0.0), 4),
            "transfers": transfers,
            "transfer_gb": round(
                METRICS.counter_value("lzy_transfer_bytes") / 1e9, 3
            ),
            "chain_dispatches": int(
                METRICS.counter_value("lzy_chain_dispatches")
            ),
            "stream_plans": int(METRICS.counter_value("lzy_stream_plans")),
            "stream_plan_ms": round(plan_stats.get("mean", 0.0) * 1e3, 3)
            if plan_stats.get("count") else None,
            # mean per-rank recv wait inside plans: ~plan_ms on the
            # critical rank means transfer-bound (expected); ~0 would
            # mean compute-bound combines
            "plan_recv_wait_ms": round(
                METRICS.timing_stats("lzy_plan_recv_wait_s").get("mean", 0.0)
                * 1e3, 3,
            ) if METRICS.timing_stats("lzy_plan_recv_wait_s").get("count")
            else None,
            "xgmi_gbps_est": xgmi_gbps_est,
            "stage_ms": stage_ms,
        },
    }
    print(json.dumps(result), flush=True)


if __name__ == "__main__":
    main()
