"""Serializer registry tests (reference: pylzy serialization tests +
custom_serializer scenario)."""
import io

import numpy as np
import pytest
import torch

from lzy_amd.serialization.registry import LzySerializerRegistry
from lzy_amd.serialization.api import Serializer
from lzy_amd.types import File


@pytest.fixture()
def reg():
    return LzySerializerRegistry()


def roundtrip(reg, obj):
    data, fmt = reg.dumps(obj)
    return reg.loads(data, fmt, type(obj))


def test_primitives(reg):
    for v in [1, 1.5, "hello", True, None, b"\x00\xff"]:
        assert roundtrip(reg, v) == v


def test_primitive_stable(reg):
    ser = reg.find_serializer_by_type(int)
    assert ser.stable()


def test_numpy(reg):
    a = np.random.rand(16, 8).astype(np.float32)
    b = roundtrip(reg, a)
    np.testing.assert_array_equal(a, b)


@pytest.mark.parametrize("dtype", [torch.float32, torch.float16, torch.bfloat16, torch.int64])
def test_tensor_roundtrip(reg, dtype):
    t = (torch.randn(33, 17) * 10).to(dtype)
    r = roundtrip(reg, t)
    assert r.dtype == dtype
    assert torch.equal(t, r)


def test_tensor_stable(reg):
    assert reg.find_serializer_by_type(torch.Tensor).stable()


def test_fallback_pickle(reg):
    class Custom:
        def __init__(self, v):
            self.v = v

    c = roundtrip(reg, Custom(3))
    assert c.v == 3
    assert not reg.find_serializer_by_type(Custom).stable()


def test_file_roundtrip(reg):
    f = File.create_tmp()
    f.write_text("file content here")
    data, fmt = reg.dumps(f)
    g = reg.loads(data, fmt, File)
    assert g.read_text() == "file content here"


def test_custom_serializer_priority(reg):
    class Point:
        def __init__(self, x, y):
            self.x, self.y = x, y

    class PointSerializer(Serializer):
        def serialize(self, obj, dest):
            dest.write(f"{obj.x},{obj.y}".encode())

        def deserialize(self, src, typ=None):
            x, y = src.read().decode().split(",")
            return Point(int(x), int(y))

        def supported_types(self):
            return Point

        def data_format(self):
            return "point_csv"

        def stable(self):
            return True

    reg.register_serializer("point", PointSerializer(), priority=0)
    assert isinstance(reg.find_serializer_by_type(Point), PointSerializer)
    p = roundtrip(reg, Point(2, 3))
    assert (p.x, p.y) == (2, 3)

    reg.unregister_serializer("point")
    assert reg.find_serializer_by_type(Point).data_format() == "pickle"


def test_duplicate_name_rejected(reg):
    with pytest.raises(ValueError):
        reg.register_serializer("primitive", reg.find_serializer_by_type(int))


def test_user_serializer_transport(reg):
    class Token:
        def __init__(self, s):
            self.s = s

    class TokenSer(Serializer):
        def serialize(self, obj, dest):
            dest.write(obj.s.encode())

        def deserialize(self, src, typ=None):
            return Token(src.read().decode())

        def supported_types(self):
            return Token

        def data_format(self):
            return "token_raw"

        def stable(self):
            return True

    reg.register_serializer("token", TokenSer(), priority=0)
    payload = reg.user_serializers_payload()

    other = LzySerializerRegistry()
    other.load_user_serializers(payload)
    assert other.find_serializer_by_data_format("token_raw") is not None


def test_mem_storage_client():
    """mem:// blobs (reference: InMemoryS3Storage test fleet)."""
    from lzy_amd.storage.api import StorageConfig, StorageRegistry
    from lzy_amd.storage.mem import reset_mem_storage

    reset_mem_storage()
    reg = StorageRegistry()
    reg.register_storage("m", StorageConfig(uri="mem://t"), default=True)
    c = reg.default_client()
    c.write_bytes("mem://t/a/b", b"payload")
    assert c.blob_exists("mem://t/a/b")
    assert c.read_bytes("mem://t/a/b") == b"payload"
    assert c.size_in_bytes("mem://t/a/b") == 7
    c.copy("mem://t/a/b", "mem://t/c")
    assert c.read_bytes("mem://t/c") == b"payload"
    reset_mem_storage()
    assert not c.blob_exists("mem://t/a/b")
