from lzy_amd.runtime.base import Runtime
from lzy_amd.runtime.local import LocalRuntime

__all__ = ["Runtime", "LocalRuntime"]
