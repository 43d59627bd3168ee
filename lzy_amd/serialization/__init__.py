from lzy_amd.serialization.api import Schema, Serializer
from lzy_amd.serialization.registry import LzySerializerRegistry

__all__ = ["Schema", "Serializer", "LzySerializerRegistry"]
