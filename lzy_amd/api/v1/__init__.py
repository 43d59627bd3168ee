"""Drop-in import surface matching the reference's ``lzy.api.v1``
(reference: pylzy/lzy/api/v1/__init__.py exports).  Everything resolves
to the MI355X-native implementations."""
from lzy_amd.core.lzy import Lzy, lzy_auth
from lzy_amd.core.op import op
from lzy_amd.core.workflow import LzyWorkflow
from lzy_amd.env.environment import LzyEnvironment
from lzy_amd.env.provisioning import GpuType, Provisioning
from lzy_amd.env.shortcuts import (
    auto_python,
    cpu,
    docker_image,
    env_vars,
    gpu,
    manual_python,
    ram,
)
from lzy_amd.runtime.base import Runtime
from lzy_amd.runtime.local import LocalRuntime
from lzy_amd.runtime.pool import GpuPoolRuntime
from lzy_amd.types import File
from lzy_amd.whiteboards.wb import whiteboard, whiteboard_

__all__ = [
    "Lzy", "lzy_auth", "op", "LzyWorkflow", "LzyEnvironment",
    "Provisioning", "GpuType", "Runtime", "LocalRuntime", "GpuPoolRuntime", "File",
    "whiteboard", "whiteboard_", "gpu", "cpu", "ram", "env_vars",
    "auto_python", "manual_python", "docker_image",
]
