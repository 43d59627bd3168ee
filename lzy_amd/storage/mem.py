"""In-memory storage client (reference: the in-memory S3 used by the
test fleet — lzy-service storage/InMemoryS3Storage.java:32).

``mem://<space>/<path>`` blobs live in a process-wide dict; useful for
tests and ephemeral runs where even local NVMe writes are unwanted.
"""
from __future__ import annotations

import shutil
import threading
from typing import BinaryIO, Dict

from lzy_amd.storage.api import StorageClient

_BLOBS: Dict[str, bytes] = {}
_LOCK = threading.Lock()


def reset_mem_storage() -> None:
    with _LOCK:
        _BLOBS.clear()


class MemStorageClient(StorageClient):
    def read(self, uri: str, dest: BinaryIO) -> None:
        with _LOCK:
            data = _BLOBS.get(uri)
        if data is None:
            raise FileNotFoundError(uri)
        dest.write(data)

    def write(self, uri: str, data: BinaryIO) -> None:
        payload = data.read()
        with _LOCK:
            _BLOBS[uri] = payload

    def blob_exists(self, uri: str) -> bool:
        with _LOCK:
            return uri in _BLOBS

    def copy(self, from_uri: str, to_uri: str) -> None:
        with _LOCK:
            if from_uri not in _BLOBS:
                raise FileNotFoundError(from_uri)
            _BLOBS[to_uri] = _BLOBS[from_uri]

    def size_in_bytes(self, uri: str) -> int:
        with _LOCK:
            if uri not in _BLOBS:
                raise FileNotFoundError(uri)
            return len(_BLOBS[uri])
