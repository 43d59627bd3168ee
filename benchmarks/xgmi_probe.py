"""xGMI p2p micro-benchmark (run under torchrun, one rank per GPU).

Measures, per pattern, the effective bandwidth of the transport layer's
grouped batch_isend_irecv:

  pairwise  — disjoint pairs (0<->1, 2<->3, ...): per-link bandwidth
  gather    — all ranks send to rank 0 in ONE group: tests link
              aggregation into a single GPU (7 links on MI355X)
  ring      — each rank sends to (r+1)%N: ring-step bandwidth

Usage:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 benchmarks/xgmi_probe.py [--mb 512] [--iters 5]

Falls back to gloo/CPU when GPUs are unavailable (logic smoke only).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def bench_pattern(name, ops_fn, nbytes, iters, rank):
    for _ in range(2):  # warmup
        works = dist.batch_isend_irecv(ops_fn()) if ops_fn() else []
        for w in works:
            w.wait()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        ops = ops_fn()
        if ops:
            for w in dist.batch_isend_irecv(ops):
                w.wait()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
    dist.barrier()
    dt = (time.perf_counter() - t0) / iters
    if rank == 0:
        print(f"{name:9s}: {nbytes / dt / 1e9:7.1f} GB/s moved/iter "
              f"({nbytes >> 20} MiB, {dt * 1e3:.2f} ms)", flush=True)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--mb", type=int, default=512)
    ap.add_argument("--iters", type=int, default=5)
    args = ap.parse_args()

    use_cuda = torch.cuda.is_available()
    backend = "nccl" if use_cuda else "gloo"
    dist.init_process_group(backend)
    rank, world = dist.get_rank(), dist.get_world_size()
    if use_cuda:
        torch.cuda.set_device(rank % torch.cuda.device_count())
    dev = torch.device("cuda") if use_cuda else torch.device("cpu")

    n = (args.mb << 20) // 2
    buf = torch.ones(n, dtype=torch.bfloat16, device=dev)
    rbuf = torch.empty(n, dtype=torch.bfloat16, device=dev)

    # pairwise disjoint
    def pairwise():
        peer = rank ^ 1
        if peer >= world:
            return []
        if rank < peer:
            return [dist.P2POp(dist.isend, buf, peer),
                    dist.P2POp(dist.irecv, rbuf, peer)]
        return [dist.P2POp(dist.irecv, rbuf, peer),
                dist.P2POp(dist.isend, buf, peer)]

    bench_pattern("pairwise", pairwise, buf.numel() * 2 * (world // 2) * 2,
                  args.iters, rank)

    # gather into rank 0 (one grouped batch of N-1 recvs)
    gbufs = [torch.empty(n, dtype=torch.bfloat16, device=dev)
             for _ in range(world - 1)] if rank == 0 else []

    def gather():
        if rank == 0:
            return [dist.P2POp(dist.irecv, gbufs[i], i + 1)
                    for i in range(world - 1)]
        return [dist.P2POp(dist.isend, buf, 0)]

    bench_pattern("gather", gather, buf.numel() * 2 * (world - 1),
                  args.iters, rank)

    # ring step
    def ring():
        nxt, prv = (rank + 1) % world, (rank - 1) % world
        if world == 1:
            return []
        return [dist.P2POp(dist.isend, buf, nxt),
                dist.P2POp(dist.irecv, rbuf, prv)]

    bench_pattern("ring", ring, buf.numel() * 2 * world, args.iters, rank)

    dist.destroy_process_group()


if __name__ == "__main__":
    main()
