"""Grid/variant sweep for the data-plane kernels (run on an MI355X).

Times cast_copy and both checksum variants over grid caps, printing GB/s.
"""
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch

import lzy_amd.ops as ops


def timeit(fn, iters=8):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main() -> None:
    lib = ops._require_native()
    n = 256 << 20
    src = torch.randn(n, device="cuda")
    dst16 = torch.empty(n, device="cuda", dtype=torch.bfloat16)

    for cap in [1024, 2048, 4096, 8192, 16384]:
        lib.lz_set_max_blocks(cap)
        dt = timeit(lambda: ops.cast_copy(src, dst16))
        print(f"cast f32->bf16 cap={cap:6d}: {(n*4+n*2)/dt/1e9:7.0f} GB/s")
    for cap in [1024, 2048, 4096, 8192, 16384]:
        lib.lz_set_max_blocks(cap)
        dt = timeit(lambda: ops.device_checksum(src, method="valu"))
        print(f"checksum valu  cap={cap:6d}: {(n*4)/dt/1e9:7.0f} GB/s")
    for cap in [1024, 2048, 4096, 8192, 16384]:
        lib.lz_set_max_blocks(cap)
        dt = timeit(lambda: ops.device_checksum(src, method="mfma"))
        print(f"checksum mfma  cap={cap:6d}: {(n*4)/dt/1e9:7.0f} GB/s")
    lib.lz_set_max_blocks(0)

    # hbm ceiling references
    dt = timeit(lambda: dst16.copy_(src))
    print(f"torch copy_ f32->bf16 ref:  {(n*4+n*2)/dt/1e9:7.0f} GB/s")
    big = torch.empty(n, device="cuda", dtype=torch.float32)
    dt = timeit(lambda: big.copy_(src))
    print(f"torch copy_ f32->f32 ref:   {(n*8)/dt/1e9:7.0f} GB/s")
    dt = timeit(lambda: float(src.sum()))
    print(f"torch sum (read 1GiB) ref:  {(n*4)/dt/1e9:7.0f} GB/s")


if __name__ == "__main__":
    main()
