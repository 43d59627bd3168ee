"""User-registered serializer drives op IO for a custom type (reference
scenario: custom_serializer)."""
from typing import BinaryIO, Optional, Type

from lzy_amd import Lzy, op
from lzy_amd.serialization.api import Serializer


class Point:
    def __init__(self, x: int, y: int):
        self.x, self.y = x, y


class PointSerializer(Serializer):
    def serialize(self, obj, dest: BinaryIO) -> None:
        print("custom serialize")
        dest.write(f"{obj.x},{obj.y}".encode())

    def deserialize(self, source: BinaryIO, typ: Optional[Type] = None):
        x, y = source.read().decode().split(",")
        return Point(int(x), int(y))

    def supported_types(self) -> Type:
        return Point

    def stable(self) -> bool:
        return True

    def data_format(self) -> str:
        return "point-csv"


@op
def shift(p: Point) -> Point:
    return Point(p.x + 1, p.y + 1)


if __name__ == "__main__":
    lzy = Lzy()
    lzy.serializer_registry.register_serializer("point", PointSerializer())
    with lzy.workflow("wf", interactive=False):
        q = shift(Point(1, 2))
        print(f"point {q.x} {q.y}")
