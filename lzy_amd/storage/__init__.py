from lzy_amd.storage.api import StorageConfig, StorageRegistry
from lzy_amd.storage.fs import FsStorageClient

__all__ = ["StorageConfig", "StorageRegistry", "FsStorageClient"]
