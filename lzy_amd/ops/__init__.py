"""HIP/CDNA4 data-plane kernels (gfx950).

``libhipops.so`` is the in-tree HIP library (lzy_amd/ops/hipops.hip):
vectorized pack/cast and the content-checksum kernels, launched on the
caller's current torch HIP stream via a thin ctypes binding (no torch C++
ABI coupling — tensors are passed as raw device pointers, torch's caching
allocator owns all memory).

On a GPU box the native path is mandatory: ops raise
NativeExtensionMissing instead of silently falling back to eager torch.
"""
from __future__ import annotations

import ctypes
import os
from typing import Optional

from lzy_amd.exceptions import NativeExtensionMissing

_LIB_PATH = os.path.join(os.path.dirname(__file__), "libhipops.so")

_lib: Optional[ctypes.CDLL] = None
_load_error: Optional[str] = None


def _try_load() -> Optional[ctypes.CDLL]:
    global _lib, _load_error
    if _lib is not None:
        return _lib
    if not os.path.exists(_LIB_PATH):
        _load_error = f"{_LIB_PATH} not built"
        return None
    try:
        # Bind to torch's HIP runtime, not the system one: torch bundles
        # its own libamdhip64.so.7 and two coexisting runtimes make our
        # launches fail with hipErrorNoDevice.  Loading torch (and its
        # runtime, RTLD_GLOBAL) first makes our DT_NEEDED resolve to the
        # same runtime torch uses.
        import torch  # noqa: F401

        torch_hip = os.path.join(
            os.path.dirname(torch.__file__), "lib", "libamdhip64.so"
        )
        if os.path.exists(torch_hip):
            ctypes.CDLL(torch_hip, mode=ctypes.RTLD_GLOBAL)
        lib = ctypes.CDLL(_LIB_PATH)
    except OSError as e:  # e.g. no ROCm runtime on a CPU-only box
        _load_error = str(e)
        return None
    lib.lz_cast_copy.restype = ctypes.c_int
    lib.lz_cast_copy.argtypes = [
        ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p, ctypes.c_int,
        ctypes.c_int64, ctypes.c_void_p,
    ]
    lib.lz_checksum.restype = ctypes.c_int
    lib.lz_checksum.argtypes = [
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
    ]
    lib.lz_checksum_mfma.restype = ctypes.c_int
    lib.lz_checksum_mfma.argtypes = [
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p,
    ]
    lib.lz_fill_pattern.restype = ctypes.c_int
    lib.lz_fill_pattern.argtypes = [
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_uint64, ctypes.c_void_p,
    ]
    lib.lz_fill_pattern_masked.restype = ctypes.c_int
    lib.lz_fill_pattern_masked.argtypes = [
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_uint64, ctypes.c_uint64,
        ctypes.c_void_p,
    ]
    lib.lz_stats.restype = ctypes.c_int
    lib.lz_stats.argtypes = [
        ctypes.c_void_p, ctypes.c_int, ctypes.c_int64, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_void_p,
    ]
    lib.lz_stats_variant.restype = ctypes.c_int
    lib.lz_stats_variant.argtypes = [
        ctypes.c_void_p, ctypes.c_int, ctypes.c_int64, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
    ]
    lib.lz_normalize_apply.restype = ctypes.c_int
    lib.lz_normalize_apply.argtypes = [
        ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p, ctypes.c_int,
        ctypes.c_int64, ctypes.c_void_p, ctypes.c_float, ctypes.c_void_p,
    ]
    lib.lz_scale_shift.restype = ctypes.c_int
    lib.lz_scale_shift.argtypes = [
        ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p, ctypes.c_int,
        ctypes.c_int64, ctypes.c_float, ctypes.c_float, ctypes.c_void_p,
    ]
    lib.lz_transpose_cast.restype = ctypes.c_int
    lib.lz_transpose_cast.argtypes = [
        ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p, ctypes.c_int,
        ctypes.c_int64, ctypes.c_int64, ctypes.c_void_p,
    ]
    lib.lz_axpby.restype = ctypes.c_int
    lib.lz_axpby.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
        ctypes.c_int, ctypes.c_int64, ctypes.c_float, ctypes.c_float,
        ctypes.c_void_p,
    ]
    lib.lz_error_name.restype = ctypes.c_char_p
    lib.lz_error_name.argtypes = [ctypes.c_int]
    lib.lz_set_max_blocks.restype = None
    lib.lz_set_max_blocks.argtypes = [ctypes.c_int]
    cap = os.environ.get("LZY_HIP_MAX_BLOCKS")
    if cap:
        lib.lz_set_max_blocks(int(cap))
    _lib = lib
    return lib


NATIVE = _try_load() is not None


def _require_native() -> ctypes.CDLL:
    lib = _try_load()
    if lib is None:
        raise NativeExtensionMissing(
            f"lzy_amd HIP ops not available ({_load_error}); build with "
            f"`python setup.py build_ext --inplace` / __graft_entry__.build() "
            f"(hipcc --offload-arch=gfx950)"
        )
    return lib


def _check(err: int) -> None:
    if err != 0:
        lib = _require_native()
        name = lib.lz_error_name(err).decode()
        raise RuntimeError(f"hipops kernel failed: {name} ({err})")


# dtype codes — must match LzDtype in hipops.hip
def _dtype_code(dtype) -> int:
    import torch

    codes = {
        torch.float32: 0,
        torch.float16: 1,
        torch.bfloat16: 2,
        torch.float8_e4m3fn: 3,
        torch.float8_e5m2: 4,
        torch.uint8: 5,
        torch.int32: 6,
        torch.int64: 7,
        torch.float64: 8,
    }
    if dtype not in codes:
        raise ValueError(f"unsupported dtype for hipops: {dtype}")
    return codes[dtype]


def _current_stream_ptr() -> int:
    import torch

    return torch.cuda.current_stream().cuda_stream


def cast_copy(src, dst) -> None:
    """dst[i] = cast(src[i]); both device-contiguous, same numel.

    Fused pack+dtype-cast on the current stream — the channel transport's
    cast-on-the-wire primitive.
    """
    lib = _require_native()
    if src.numel() != dst.numel():
        raise ValueError("cast_copy: numel mismatch")
    if not (src.is_cuda and dst.is_cuda):
        raise ValueError("cast_copy: device tensors required")
    s = src.detach().contiguous()
    _check(
        lib.lz_cast_copy(
            ctypes.c_void_p(s.data_ptr()),
            _dtype_code(s.dtype),
            ctypes.c_void_p(dst.data_ptr()),
            _dtype_code(dst.dtype),
            s.numel(),
            ctypes.c_void_p(_current_stream_ptr()),
        )
    )


def device_checksum(t, method: str = "auto") -> int:
    """64-bit content hash of a device tensor, computed on-GPU (no PCIe
    round-trip for the data; 8 bytes come back).

    method: "mfma" routes the mixing through the i8 matrix cores (the
    default for buffers >= 64 KiB — VALU stays free, bandwidth-bound),
    "valu" is the splitmix64 kernel, "auto" picks by size.  The two
    digests are different hash functions (both stable across runs).
    """
    import torch

    lib = _require_native()
    if not t.is_cuda:
        raise ValueError("device_checksum requires a device tensor")
    flat = t.detach().contiguous().view(torch.uint8) if t.dtype != torch.uint8 \
        else t.detach().contiguous()
    nbytes = flat.numel() * flat.element_size()
    if method == "auto":
        method = "mfma" if nbytes >= (64 << 10) else "valu"
    fn = lib.lz_checksum_mfma if method == "mfma" else lib.lz_checksum
    out = torch.zeros(1, dtype=torch.int64, device=t.device)
    _check(
        fn(
            ctypes.c_void_p(flat.data_ptr()),
            nbytes,
            ctypes.c_void_p(out.data_ptr()),
            ctypes.c_void_p(_current_stream_ptr()),
        )
    )
    return int(out.item()) & 0xFFFFFFFFFFFFFFFF


def stats(t, use_abs: bool = False):
    """One-pass (sum, sumsq) — or (sum|x|, sumsq) — of a device tensor.
    Returns a device float64[2] tensor; no host sync."""
    import torch

    lib = _require_native()
    if not t.is_cuda:
        raise ValueError("stats requires a device tensor")
    flat = t.detach().contiguous()
    out = torch.empty(2, dtype=torch.float64, device=t.device)
    _check(
        lib.lz_stats(
            ctypes.c_void_p(flat.data_ptr()),
            _dtype_code(flat.dtype),
            flat.numel(),
            1 if use_abs else 0,
            ctypes.c_void_p(out.data_ptr()),
            ctypes.c_void_p(_current_stream_ptr()),
        )
    )
    return out


def stats_variant(t, variant: int, use_abs: bool = False):
    """Sweep-only: run a (V, U) load-shape variant of the stats kernel
    (benchmarks/kernel_sweep.py picks the default)."""
    import torch

    lib = _require_native()
    flat = t.detach().contiguous()
    out = torch.empty(2, dtype=torch.float64, device=t.device)
    _check(
        lib.lz_stats_variant(
            ctypes.c_void_p(flat.data_ptr()),
            _dtype_code(flat.dtype),
            flat.numel(),
            1 if use_abs else 0,
            ctypes.c_void_p(out.data_ptr()),
            ctypes.c_void_p(_current_stream_ptr()),
            int(variant),
        )
    )
    return out


def abs_mean(t) -> float:
    """Fused mean(|x|) — one read of the buffer, 8 bytes back to host.
    (torch chain `t.float().abs().mean()` moves ~9x the bytes.)"""
    s = stats(t, use_abs=True)
    return float(s[0].item()) / max(1, t.numel())


def mean_std(t, unbiased: bool = False):
    """(mean, std) from the one-pass stats kernel; two scalars back."""
    import math

    s = stats(t, use_abs=False)
    n = t.numel()
    vals = s.cpu()
    mean = float(vals[0]) / n
    var = float(vals[1]) / n - mean * mean
    if unbiased and n > 1:
        var *= n / (n - 1)
    return mean, math.sqrt(max(var, 0.0))


def normalize(src, dst=None, eps: float = 1e-6):
    """dst = (src - mean(src)) / (std(src) + ~eps), fused two-pass on
    device (stats kernel + apply kernel, stats handed over in HBM — no
    host round-trip).  dst defaults to a new tensor of src's dtype."""
    import torch

    lib = _require_native()
    if not src.is_cuda:
        raise ValueError("normalize requires a device tensor")
    s = src.detach().contiguous()
    if dst is None:
        dst = torch.empty_like(s)
    st = stats(s, use_abs=False)
    _check(
        lib.lz_normalize_apply(
            ctypes.c_void_p(s.data_ptr()),
            _dtype_code(s.dtype),
            ctypes.c_void_p(dst.data_ptr()),
            _dtype_code(dst.dtype),
            s.numel(),
            ctypes.c_void_p(st.data_ptr()),
            ctypes.c_float(eps),
            ctypes.c_void_p(_current_stream_ptr()),
        )
    )
    return dst


def scale_shift(src, a: float, b: float, dst=None):
    """dst = src * a + b (fused single-FMA elementwise; in-place when
    dst is src)."""
    import torch

    lib = _require_native()
    if not src.is_cuda:
        raise ValueError("scale_shift requires a device tensor")
    s = src.detach().contiguous()
    if dst is None:
        dst = torch.empty_like(s)
    _check(
        lib.lz_scale_shift(
            ctypes.c_void_p(s.data_ptr()),
            _dtype_code(s.dtype),
            ctypes.c_void_p(dst.data_ptr()),
            _dtype_code(dst.dtype),
            s.numel(),
            ctypes.c_float(a),
            ctypes.c_float(b),
            ctypes.c_void_p(_current_stream_ptr()),
        )
    )
    return dst


def axpby(a, b, alpha: float = 1.0, beta: float = 1.0, dst=None):
    """dst = alpha*a + beta*b, fused (the tree-merge primitive: one read
    of each input, one write, vs 4+ passes for the torch float() chain)."""
    import torch

    lib = _require_native()
    if not (a.is_cuda and b.is_cuda):
        raise ValueError("axpby requires device tensors")
    if a.numel() != b.numel() or a.dtype != b.dtype:
        raise ValueError("axpby: shape/dtype mismatch")
    x = a.detach().contiguous()
    y = b.detach().contiguous()
    if dst is None:
        dst = torch.empty_like(x)
    _check(
        lib.lz_axpby(
            ctypes.c_void_p(x.data_ptr()),
            ctypes.c_void_p(y.data_ptr()),
            _dtype_code(x.dtype),
            ctypes.c_void_p(dst.data_ptr()),
            _dtype_code(dst.dtype),
            x.numel(),
            ctypes.c_float(alpha),
            ctypes.c_float(beta),
            ctypes.c_void_p(_current_stream_ptr()),
        )
    )
    return dst


def transpose_cast(src, dst=None, dtype=None):
    """dst = src.T (2-D), optionally casting — LDS-staged 64x64 tiles so
    both the gather and the scatter side stay coalesced (the pack path
    for transposed tensors; a naive strided copy runs at ~1/8 HBM rate)."""
    import torch

    lib = _require_native()
    if not src.is_cuda or src.dim() != 2:
        raise ValueError("transpose_cast requires a 2-D device tensor")
    s = src.detach().contiguous()
    rows, cols = s.shape
    out_dtype = dtype or s.dtype
    if dst is None:
        dst = torch.empty(cols, rows, dtype=out_dtype, device=s.device)
    _check(
        lib.lz_transpose_cast(
            ctypes.c_void_p(s.data_ptr()),
            _dtype_code(s.dtype),
            ctypes.c_void_p(dst.data_ptr()),
            _dtype_code(dst.dtype),
            rows,
            cols,
            ctypes.c_void_p(_current_stream_ptr()),
        )
    )
    return dst


def fill_pattern(t, seed: int = 0, mask16: int = 0) -> None:
    """Deterministic device-side fill (tests / synthetic data).

    ``mask16``: AND every 16-bit lane with this mask in the same pass —
    e.g. 0x3FFF makes a bf16 buffer finite positive synthetic data
    without a second read+write over HBM."""
    lib = _require_native()
    flat = t.detach().contiguous()
    nbytes = flat.numel() * flat.element_size()
    if mask16:
        _check(
            lib.lz_fill_pattern_masked(
                ctypes.c_void_p(flat.data_ptr()),
                nbytes,
                ctypes.c_uint64(seed),
                ctypes.c_uint64(mask16),
                ctypes.c_void_p(_current_stream_ptr()),
            )
        )
        return
    _check(
        lib.lz_fill_pattern(
            ctypes.c_void_p(flat.data_ptr()),
            nbytes,
            ctypes.c_uint64(seed),
            ctypes.c_void_p(_current_stream_ptr()),
        )
    )
