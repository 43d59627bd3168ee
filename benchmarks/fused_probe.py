"""Probe target for rocprofv3 PMC capture of the fused data-plane kernels.

Usage: python benchmarks/fused_probe.py [stats|abs_mean|normalize|scale_shift|axpby] [iters]
Runs the selected fused kernel on 512 MiB bf16 device buffers and prints
un-profiled effective bandwidth.
"""
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch

import lzy_amd.ops as ops


def main() -> None:
    which = sys.argv[1] if len(sys.argv) > 1 else "normalize"
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 10
    n = 256 << 20
    t = torch.empty(n, device="cuda", dtype=torch.bfloat16)
    t.normal_()
    nbytes = n * 2
    out = torch.empty_like(t)

    if which == "stats":
        fn = lambda: ops.stats(t)  # noqa: E731
        moved = nbytes
    elif which == "abs_mean":
        fn = lambda: ops.abs_mean(t)  # noqa: E731
        moved = nbytes
    elif which == "normalize":
        fn = lambda: ops.normalize(t, dst=out)  # noqa: E731
        moved = 3 * nbytes
    elif which == "scale_shift":
        fn = lambda: ops.scale_shift(t, 1.001, 0.125, dst=out)  # noqa: E731
        moved = 2 * nbytes
    elif which == "axpby":
        b = torch.empty_like(t)
        b.normal_()
        fn = lambda: ops.axpby(t, b, 0.5, 0.5, dst=out)  # noqa: E731
        moved = 3 * nbytes
    else:
        raise SystemExit(f"unknown probe {which}")

    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"{which}: {moved / dt / 1e9:.0f} GB/s effective ({iters}x)")


if __name__ == "__main__":
    main()
