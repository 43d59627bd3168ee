"""Serializer interface.

Same capability as the external ``serialzy`` package the reference SDK wraps
(reference: pylzy/lzy/serialization/registry.py:20), re-designed for a
GPU-resident data plane:

  * serializers write to / read from binary streams (spill tier, files);
  * a serializer declares ``stable()`` — whether its byte format is portable
    across environments (whiteboard fields require stable formats, matching
    reference api/v1/whiteboards.py:100-110);
  * torch tensors have a *device-aware* path: ``TensorSerializer`` writes a
    raw header + contiguous bytes, and the data plane may skip serialization
    entirely for device→device channel moves (lzy_amd/channels).
"""
from __future__ import annotations

import abc
from dataclasses import dataclass, field
from typing import Any, BinaryIO, Callable, Dict, Optional, Type, Union


@dataclass(frozen=True)
class Schema:
    """Descriptor of serialized data (reference analogue: serialzy Schema)."""

    data_format: str
    schema_format: str
    schema_content: str = ""
    meta: Dict[str, str] = field(default_factory=dict)


class Serializer(abc.ABC):
    @abc.abstractmethod
    def serialize(self, obj: Any, dest: BinaryIO) -> None: ...

    @abc.abstractmethod
    def deserialize(self, source: BinaryIO, typ: Optional[Type] = None) -> Any: ...

    @abc.abstractmethod
    def supported_types(self) -> Union[Type, Callable[[Type], bool]]: ...

    @abc.abstractmethod
    def data_format(self) -> str: ...

    def available(self) -> bool:
        return True

    def stable(self) -> bool:
        """True if the format is portable across python environments."""
        return False

    def requirements(self) -> str:
        return ""

    def meta(self) -> Dict[str, str]:
        return {}

    def schema(self, typ: Type) -> Schema:
        return Schema(
            data_format=self.data_format(),
            schema_format="lzy_amd_type_name",
            schema_content=f"{typ.__module__}.{getattr(typ, '__qualname__', typ)}",
            meta=self.meta(),
        )

    def resolve(self, schema: Schema) -> Type:
        """Best-effort type resolution from a schema (used by whiteboards)."""
        mod, _, name = schema.schema_content.rpartition(".")
        try:
            import importlib

            return getattr(importlib.import_module(mod), name)
        except Exception:
            return object

    def matches(self, typ: Type) -> bool:
        st = self.supported_types()
        if isinstance(st, type):
            try:
                return issubclass(typ, st)
            except TypeError:
                return False
        return bool(st(typ))
