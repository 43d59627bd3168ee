"""Storage abstraction + registry.

Reference capability (pylzy/lzy/storage/api.py:10-56, storage/registry.py:8):
pluggable storage backends keyed by name with one default.  The MI355X
re-design keeps the durable tier pluggable (file:// on local NVMe by
default — there is no S3 on a single node and no network in the target
environment) while the *hot* path never touches it: values live in the
in-process result store (HBM / host RAM) and are persisted lazily.
"""
from __future__ import annotations

import abc
from dataclasses import dataclass
from typing import BinaryIO, Dict, Optional, Tuple


@dataclass(frozen=True)
class StorageConfig:
    uri: str  # e.g. "file:///tmp/lzy_amd_storage"


class StorageClient(abc.ABC):
    @abc.abstractmethod
    def read(self, uri: str, dest: BinaryIO) -> None: ...

    @abc.abstractmethod
    def write(self, uri: str, data: BinaryIO) -> None: ...

    @abc.abstractmethod
    def blob_exists(self, uri: str) -> bool: ...

    @abc.abstractmethod
    def copy(self, from_uri: str, to_uri: str) -> None: ...

    @abc.abstractmethod
    def size_in_bytes(self, uri: str) -> int: ...

    def read_bytes(self, uri: str) -> bytes:
        import io

        buf = io.BytesIO()
        self.read(uri, buf)
        return buf.getvalue()

    def write_bytes(self, uri: str, data: bytes) -> None:
        import io

        self.write(uri, io.BytesIO(data))


class StorageRegistry:
    """Named storage configs with a default (reference DefaultStorageRegistry)."""

    def __init__(self) -> None:
        self._configs: Dict[str, StorageConfig] = {}
        self._clients: Dict[str, StorageClient] = {}
        self._default: Optional[str] = None

    def register_storage(self, name: str, config: StorageConfig, default: bool = False) -> None:
        if config.uri.startswith("file://"):
            from lzy_amd.storage.fs import FsStorageClient

            client: "StorageClient" = FsStorageClient()
        elif config.uri.startswith("mem://"):
            from lzy_amd.storage.mem import MemStorageClient

            client = MemStorageClient()
        else:
            raise ValueError(
                f"Unsupported storage scheme for single-node runtime: {config.uri}"
            )
        self._configs[name] = config
        self._clients[name] = client
        if default or self._default is None:
            self._default = name

    def unregister_storage(self, name: str) -> None:
        self._configs.pop(name, None)
        self._clients.pop(name, None)
        if self._default == name:
            self._default = next(iter(self._configs), None)

    def config(self, name: str) -> Optional[StorageConfig]:
        return self._configs.get(name)

    def client(self, name: str) -> Optional[StorageClient]:
        return self._clients.get(name)

    def default_storage_name(self) -> Optional[str]:
        return self._default

    def default_config(self) -> Optional[StorageConfig]:
        return self._configs.get(self._default) if self._default else None

    def default_client(self) -> Optional[StorageClient]:
        return self._clients.get(self._default) if self._default else None
