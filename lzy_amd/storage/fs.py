"""file:// storage client (durable tier on local NVMe).

Reference analogue: pylzy/lzy/storage/async_/fs.py:10.  Writes are
atomic (tmp file + rename) so a crashed op never leaves a half-written
blob that the result cache would mistake for a completed output.
"""
from __future__ import annotations

import os
import shutil
import tempfile
from pathlib import Path
from typing import BinaryIO
from urllib.parse import urlparse

from lzy_amd.storage.api import StorageClient

_CHUNK = 16 << 20


def uri_to_path(uri: str) -> Path:
    parsed = urlparse(uri)
    if parsed.scheme != "file":
        raise ValueError(f"FsStorageClient supports file:// URIs only, got {uri}")
    return Path(parsed.path)


class FsStorageClient(StorageClient):
    def read(self, uri: str, dest: BinaryIO) -> None:
        with open(uri_to_path(uri), "rb") as f:
            shutil.copyfileobj(f, dest, _CHUNK)

    def write(self, uri: str, data: BinaryIO) -> None:
        """Atomic (tmp + rename), retried on transient OSErrors when the
        source is rewindable (reference: util-db withRetries around every
        persist — DbHelper.java:25-33)."""
        from lzy_amd.utils.retry import with_retries

        start = data.tell() if data.seekable() else None

        def go() -> None:
            if start is not None:
                data.seek(start)
            path = uri_to_path(uri)
            path.parent.mkdir(parents=True, exist_ok=True)
            fd, tmp = tempfile.mkstemp(dir=path.parent, prefix=".lzy_tmp_")
            try:
                with os.fdopen(fd, "wb") as f:
                    shutil.copyfileobj(data, f, _CHUNK)
                os.replace(tmp, path)
            except BaseException:
                try:
                    os.unlink(tmp)
                except OSError:
                    pass
                raise

        with_retries(
            go,
            attempts=5 if start is not None else 1,
            retry_on=(OSError,),
            what=f"write {uri}",
        )

    def blob_exists(self, uri: str) -> bool:
        return uri_to_path(uri).is_file()

    def copy(self, from_uri: str, to_uri: str) -> None:
        dst = uri_to_path(to_uri)
        dst.parent.mkdir(parents=True, exist_ok=True)
        shutil.copyfile(uri_to_path(from_uri), dst)

    def size_in_bytes(self, uri: str) -> int:
        return uri_to_path(uri).stat().st_size
