"""Concrete serializers.

Reference capability map (not a port):
  * primitives  -> stable JSON         (serialzy primitive serializers)
  * torch.Tensor-> stable raw format   (device-aware; header + contiguous bytes)
  * numpy       -> stable .npy
  * File        -> raw content stream  (reference: pylzy/lzy/serialization/file.py)
  * fallback    -> cloudpickle, UNSTABLE (reference registry default)
"""
from __future__ import annotations

import json
import struct
from typing import Any, BinaryIO, Callable, Optional, Type, Union

from lzy_amd.serialization.api import Serializer
from lzy_amd.types import File

_PRIMITIVES = (int, float, str, bool, bytes, type(None))

_MAGIC_TENSOR = b"LZAT\x01"
_CHUNK = 16 << 20  # 16 MiB streaming chunks


class PrimitiveSerializer(Serializer):
    def serialize(self, obj: Any, dest: BinaryIO) -> None:
        if isinstance(obj, bytes):
            payload = {"t": "bytes", "v": obj.hex()}
        else:
            payload = {"t": type(obj).__name__, "v": obj}
        dest.write(json.dumps(payload).encode("utf-8"))

    def deserialize(self, source: BinaryIO, typ: Optional[Type] = None) -> Any:
        payload = json.loads(source.read().decode("utf-8"))
        if payload["t"] == "bytes":
            return bytes.fromhex(payload["v"])
        return payload["v"]

    def supported_types(self) -> Union[Type, Callable[[Type], bool]]:
        return lambda t: t in _PRIMITIVES

    def data_format(self) -> str:
        return "json_primitive"

    def stable(self) -> bool:
        return True


class TensorSerializer(Serializer):
    """Raw torch tensor format: magic | dtype | ndim | shape | bytes.

    Device tensors are staged through CPU pinned memory on serialize; the
    fast path (device→device channels) never reaches this serializer —
    see lzy_amd/channels.  Stable: readable with torch.frombuffer anywhere.
    """

    def serialize(self, obj: Any, dest: BinaryIO) -> None:
        import torch

        t = obj.detach()
        if t.device.type != "cpu":
            t = t.cpu()
        t = t.contiguous()
        dtype_name = str(t.dtype).replace("torch.", "")
        shape = tuple(t.shape)
        dest.write(_MAGIC_TENSOR)
        head = json.dumps({"dtype": dtype_name, "shape": shape}).encode()
        dest.write(struct.pack("<I", len(head)))
        dest.write(head)
        storage_bytes = t.numpy(force=True).tobytes() if t.dtype not in (
            torch.bfloat16, torch.float16, torch.float8_e4m3fn, torch.float8_e5m2,
        ) else t.view(torch.uint8).numpy(force=True).tobytes()
        dest.write(storage_bytes)

    def deserialize(self, source: BinaryIO, typ: Optional[Type] = None) -> Any:
        import torch

        magic = source.read(len(_MAGIC_TENSOR))
        if magic != _MAGIC_TENSOR:
            raise ValueError("bad tensor stream magic")
        (hlen,) = struct.unpack("<I", source.read(4))
        head = json.loads(source.read(hlen).decode())
        dtype = getattr(torch, head["dtype"])
        shape = tuple(head["shape"])
        raw = source.read()
        n = 1
        for s in shape:
            n *= s
        if n == 0:  # frombuffer rejects empty buffers
            return torch.empty(shape, dtype=dtype)
        buf = bytearray(raw)  # writable for frombuffer
        t = torch.frombuffer(buf, dtype=torch.uint8)[: n * dtype.itemsize].view(dtype)
        return t.reshape(shape).clone()

    def supported_types(self) -> Union[Type, Callable[[Type], bool]]:
        import torch

        return torch.Tensor

    def data_format(self) -> str:
        return "lzy_amd_raw_tensor"

    def stable(self) -> bool:
        return True

    def meta(self):
        import torch

        return {"torch": torch.__version__}


class NumpySerializer(Serializer):
    def serialize(self, obj: Any, dest: BinaryIO) -> None:
        import numpy as np

        np.lib.format.write_array(dest, np.ascontiguousarray(obj), allow_pickle=False)

    def deserialize(self, source: BinaryIO, typ: Optional[Type] = None) -> Any:
        import numpy as np

        return np.lib.format.read_array(source, allow_pickle=False)

    def supported_types(self) -> Union[Type, Callable[[Type], bool]]:
        import numpy as np

        return np.ndarray

    def data_format(self) -> str:
        return "npy"

    def stable(self) -> bool:
        return True


class FileSerializer(Serializer):
    def serialize(self, obj: Any, dest: BinaryIO) -> None:
        with open(obj.path, "rb") as f:
            while True:
                chunk = f.read(_CHUNK)
                if not chunk:
                    break
                dest.write(chunk)

    def deserialize(self, source: BinaryIO, typ: Optional[Type] = None) -> Any:
        out = File.create_tmp()
        with open(out.path, "wb") as f:
            while True:
                chunk = source.read(_CHUNK)
                if not chunk:
                    break
                f.write(chunk)
        return out

    def supported_types(self) -> Union[Type, Callable[[Type], bool]]:
        return File

    def data_format(self) -> str:
        return "raw_file"

    def stable(self) -> bool:
        return True


class CloudpickleSerializer(Serializer):
    """Default fallback; NOT stable across environments (reference parity:
    cloudpickle is the unstable default there too)."""

    def serialize(self, obj: Any, dest: BinaryIO) -> None:
        import cloudpickle

        cloudpickle.dump(obj, dest)

    def deserialize(self, source: BinaryIO, typ: Optional[Type] = None) -> Any:
        import cloudpickle

        return cloudpickle.load(source)

    def supported_types(self) -> Union[Type, Callable[[Type], bool]]:
        return lambda t: True

    def data_format(self) -> str:
        return "pickle"

    def stable(self) -> bool:
        return False

    def requirements(self) -> str:
        return "cloudpickle"
