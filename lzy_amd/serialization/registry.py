"""Serializer registry.

Capability parity with reference LzySerializerRegistry
(pylzy/lzy/serialization/registry.py:20): lookup by type (most-specific,
priority-ordered), lookup by data format, user registration of custom
serializers (scenario ``custom_serializer``), and serialization of the
registry's *imports* so a worker process can reconstruct user serializers
(we transport user serializers with cloudpickle instead of pip imports —
workers share the node, no env sync needed).
"""
from __future__ import annotations

import io
from typing import Any, Dict, List, Optional, Tuple, Type

from lzy_amd.exceptions import SerializationError
from lzy_amd.serialization.api import Serializer
from lzy_amd.serialization.serializers import (
    CloudpickleSerializer,
    FileSerializer,
    NumpySerializer,
    PrimitiveSerializer,
    TensorSerializer,
)


class LzySerializerRegistry:
    def __init__(self) -> None:
        # (priority, name, serializer); lower priority wins ties last —
        # user serializers are prepended.
        self._entries: List[Tuple[int, str, Serializer]] = []
        self._register_defaults()

    def _register_defaults(self) -> None:
        self.register_serializer("primitive", PrimitiveSerializer(), priority=10)
        try:
            self.register_serializer("tensor", TensorSerializer(), priority=10)
        except Exception:
            pass
        try:
            self.register_serializer("numpy", NumpySerializer(), priority=10)
        except Exception:
            pass
        self.register_serializer("file", FileSerializer(), priority=10)
        self.register_serializer("cloudpickle", CloudpickleSerializer(), priority=1_000_000)

    def register_serializer(self, name: str, serializer: Serializer, priority: int = 0) -> None:
        if any(n == name for _, n, _ in self._entries):
            raise ValueError(f"Serializer named {name!r} already registered")
        self._entries.append((priority, name, serializer))
        self._entries.sort(key=lambda e: e[0])

    def unregister_serializer(self, name: str) -> None:
        self._entries = [e for e in self._entries if e[1] != name]

    def find_serializer_by_type(self, typ: Type) -> Optional[Serializer]:
        for _, _, ser in self._entries:
            if ser.available() and ser.matches(typ):
                return ser
        return None

    def find_serializer_by_data_format(self, fmt: str) -> Optional[Serializer]:
        for _, _, ser in self._entries:
            if ser.data_format() == fmt:
                return ser
        return None

    def serializer_name(self, ser: Serializer) -> Optional[str]:
        for _, n, s in self._entries:
            if s is ser:
                return n
        return None

    # -- convenience byte-level API used by snapshot/store ------------------

    def dumps(self, obj: Any) -> Tuple[bytes, str]:
        ser = self.find_serializer_by_type(type(obj))
        if ser is None:
            raise SerializationError(f"No serializer for type {type(obj)}")
        buf = io.BytesIO()
        ser.serialize(obj, buf)
        return buf.getvalue(), ser.data_format()

    def loads(self, data: bytes, fmt: str, typ: Optional[Type] = None) -> Any:
        ser = self.find_serializer_by_data_format(fmt)
        if ser is None:
            raise SerializationError(f"No serializer for data format {fmt!r}")
        return ser.deserialize(io.BytesIO(data), typ)

    # -- worker transport ---------------------------------------------------

    def user_serializers_payload(self) -> bytes:
        """Pickle user-registered serializers for worker-side registration."""
        import cloudpickle

        user = [(p, n, s) for p, n, s in self._entries if n not in {
            "primitive", "tensor", "numpy", "file", "cloudpickle"}]
        return cloudpickle.dumps(user)

    def load_user_serializers(self, payload: bytes) -> None:
        import cloudpickle

        for p, n, s in cloudpickle.loads(payload):
            if not any(n == name for _, name, _ in self._entries):
                self.register_serializer(n, s, priority=p)
