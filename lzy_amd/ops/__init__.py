"""HIP/CDNA4 data-plane kernels (gfx950).

``_hipops`` is the in-tree HIP extension: vectorized pack/unpack with
dtype cast, MFMA-based tensor checksum for content addressing.  On a GPU
box the native path is mandatory — ops raise NativeExtensionMissing
instead of silently falling back to eager torch.
"""
from __future__ import annotations

from typing import Optional

import torch

from lzy_amd.exceptions import NativeExtensionMissing

try:
    from lzy_amd.ops import _hipops  # type: ignore[attr-defined]

    NATIVE = True
except ImportError:
    _hipops = None  # type: ignore[assignment]
    NATIVE = False


def _require_native() -> None:
    if _hipops is None:
        raise NativeExtensionMissing(
            "lzy_amd.ops._hipops is not built; run `python setup.py "
            "build_ext --inplace` (or __graft_entry__.build()) with "
            "PYTORCH_ROCM_ARCH=gfx950"
        )


def device_checksum(t: torch.Tensor) -> int:
    """64-bit content hash of a device tensor, computed on-GPU."""
    _require_native()
    if not t.is_cuda:
        raise ValueError("device_checksum requires a device tensor")
    return _hipops.checksum(t.detach().contiguous().view(-1).view(torch.uint8))


def cast_copy(src: torch.Tensor, dst: torch.Tensor) -> None:
    """Fused pack+dtype-cast: dst[i] = cast(src[i]); both contiguous, same
    element count, on the same device."""
    _require_native()
    _hipops.cast_copy(src.detach().contiguous(), dst)
