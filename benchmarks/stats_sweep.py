"""Sweep (V, U) load-shape variants of the stats reduction kernel on a
1 GiB buffer (bf16 and f32), plus grid caps.  The winner becomes the
default template parameters in hipops.hip.

Usage: python benchmarks/stats_sweep.py [iters]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from lzy_amd import ops

VARIANTS_16 = {0: "V16 U4 (cur)", 1: "V16 U8", 2: "V32 U2", 3: "V32 U4",
               4: "V8 U8", 5: "V16 U2", 6: "V8 U4",
               7: "V16 U4 NT", 8: "V8 U8 NT", 9: "V16 U2 NT"}
VARIANTS_32 = {0: "V8 U4 (cur)", 1: "V8 U8", 2: "V16 U2", 3: "V16 U4",
               4: "V4 U8", 5: "V8 U2", 6: "V4 U4",
               7: "V8 U4 NT", 8: "V16 U4 NT", 9: "V8 U2 NT"}


def bw(t, fn, iters):
    for _ in range(2):
        fn(t)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn(t)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return t.numel() * t.element_size() / dt / 1e9


def main() -> None:
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 8
    assert torch.cuda.is_available() and ops.NATIVE
    lib = ops._try_load()
    for dtype, names in ((torch.bfloat16, VARIANTS_16),
                         (torch.float32, VARIANTS_32)):
        n = (1 << 30) // torch.empty(0, dtype=dtype).element_size()
        t = torch.randn(n, dtype=torch.float32, device="cuda").to(dtype)
        ref = ops.stats(t).cpu()
        print(f"\n== stats 1 GiB {dtype} ==")
        for cap in (1024, 2048):
            lib.lz_set_max_blocks(cap)
            for v, name in names.items():
                got = ops.stats_variant(t, v).cpu()
                rel = abs(float(got[0] - ref[0])) / max(1.0, abs(float(ref[0])))
                gbs = bw(t, lambda x: ops.stats_variant(x, v), iters)
                tag = "OK" if rel < 1e-3 else f"MISMATCH {rel:.2e}"
                print(f"cap={cap} {name:14s} {gbs:7.0f} GB/s  [{tag}]")
        lib.lz_set_max_blocks(1024)
    print("\nreference reductions:")
    t = torch.randn((1 << 29,), device="cuda", dtype=torch.float32).to(torch.bfloat16)
    print(f"checksum_mfma  {bw(t, lambda x: ops.device_checksum(x), iters):7.0f} GB/s")
    print(f"torch.sum      {bw(t, lambda x: x.sum(), iters):7.0f} GB/s")


if __name__ == "__main__":
    main()
