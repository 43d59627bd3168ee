"""Probe target for rocprofv3 PMC capture of the checksum kernels.

Usage: python benchmarks/checksum_probe.py [valu|mfma] [iters]
Runs the selected checksum kernel on a 1 GiB device buffer.
"""
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch

from lzy_amd.ops import device_checksum


def main() -> None:
    method = sys.argv[1] if len(sys.argv) > 1 else "mfma"
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 10
    t = torch.randn(256 << 20, device="cuda")  # 1 GiB
    device_checksum(t, method=method)  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        h = device_checksum(t, method=method)
    dt = (time.perf_counter() - t0) / iters
    print(f"{method}: {t.numel() * 4 / dt / 1e9:.0f} GB/s digest={h:016x}")


if __name__ == "__main__":
    main()
