"""User-facing value types.

``File`` mirrors the reference's file passing capability (the reference uses
serialzy's File serializer; scenario ``pylzy/tests/scenarios/file_test``):
an @op can take and return files; content travels through the data plane.
"""
from __future__ import annotations

import os
import shutil
import tempfile
from pathlib import Path
from typing import Union


class File:
    """A file handle whose *content* is the value."""

    def __init__(self, path: Union[str, os.PathLike]):
        self.path = Path(path)

    @classmethod
    def create_tmp(cls, suffix: str = "") -> "File":
        fd, p = tempfile.mkstemp(suffix=suffix)
        os.close(fd)
        return cls(p)

    def exists(self) -> bool:
        return self.path.exists()

    def read_text(self, encoding: str = "utf-8") -> str:
        return self.path.read_text(encoding)

    def write_text(self, text: str, encoding: str = "utf-8") -> None:
        self.path.write_text(text, encoding)

    def read_bytes(self) -> bytes:
        return self.path.read_bytes()

    def write_bytes(self, data: bytes) -> None:
        self.path.write_bytes(data)

    def copy_to(self, dst: Union[str, os.PathLike]) -> None:
        shutil.copyfile(self.path, dst)

    def open(self, mode: str = "r", **kwargs):
        """File-object access (reference tutorial:
        ``with result.open("r") as f: ...``)."""
        return open(self.path, mode, **kwargs)

    def __repr__(self) -> str:
        return f"File({self.path})"

    def __eq__(self, other) -> bool:
        return isinstance(other, File) and self.path == other.path

    def __hash__(self) -> int:
        return hash(self.path)
