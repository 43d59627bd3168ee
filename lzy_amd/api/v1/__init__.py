"""Drop-in import surface matching the reference's ``lzy.api.v1``
(reference: pylzy/lzy/api/v1/__init__.py exports).  Everything resolves
to the MI355X-native implementations."""
from lzy_amd.core.lzy import Lzy, lzy_auth
from lzy_amd.core.op import op
from lzy_amd.core.workflow import LzyWorkflow
from lzy_amd.env.compat import (
    AutoPythonEnv,
    DockerContainer,
    DockerPullPolicy,
    ManualPythonEnv,
    NoContainer,
)
from lzy_amd.env.environment import LzyEnvironment
from lzy_amd.env.provisioning import (
    Any as AnyProvisioning,
    GpuType,
    Provisioning,
    maximum_score_function,
    minimum_score_function,
)
from lzy_amd.env import score
from lzy_amd.env.shortcuts import (
    auto_python,
    auto_python_env,
    cpu,
    docker_container,
    docker_image,
    env_vars,
    gpu,
    manual_python,
    manual_python_env,
    no_container,
    provisioning,
    ram,
)
from lzy_amd.runtime.base import Runtime
from lzy_amd.runtime.local import LocalRuntime
from lzy_amd.runtime.pool import GpuPoolRuntime

# Reference-named alias: the "remote" runtime of this node IS the GPU
# pool (reference: lzy/api/v1/remote/runtime.py RemoteRuntime).
RemoteRuntime = GpuPoolRuntime
from lzy_amd.types import File
from lzy_amd.whiteboards.wb import whiteboard, whiteboard_

__all__ = [
    "Lzy", "lzy_auth", "op", "LzyWorkflow", "LzyEnvironment",
    "Provisioning", "GpuType", "Runtime", "LocalRuntime", "GpuPoolRuntime", "RemoteRuntime", "File",
    "whiteboard", "whiteboard_", "gpu", "cpu", "ram", "env_vars",
    "auto_python", "manual_python", "docker_image",
    "AnyProvisioning", "DockerContainer", "DockerPullPolicy",
    "NoContainer", "AutoPythonEnv", "ManualPythonEnv",
    "maximum_score_function", "minimum_score_function",
    "provisioning", "docker_container", "no_container",
    "auto_python_env", "manual_python_env", "score",
]
