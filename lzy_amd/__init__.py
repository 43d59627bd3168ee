"""lzy_amd — MI355X-native ML-workflow runtime.

A brand-new single-node framework with the capabilities of lambdazy/lzy
(reference mounted at /root/reference): the pylzy-style ``@op`` /
``Lzy().workflow()`` / whiteboard API on top of an in-process DAG
scheduler that places ops on the 8 MI355X GPUs of one node, zero-copy
device-tensor channels over xGMI, HIP/CDNA4 kernels for the data-plane
pack/cast/checksum ops, and RCCL (torch.distributed "nccl") for
multi-GPU ops.
"""
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
from lzy_amd.proxy import (
    is_lzy_proxy,
    materialize,
    materialized,
)
from lzy_amd.types import File
from lzy_amd.whiteboards.wb import whiteboard_, whiteboard

__version__ = "0.1.0"

__all__ = [
    "Lzy",
    "lzy_auth",
    "op",
    "LzyWorkflow",
    "LzyEnvironment",
    "Provisioning",
    "GpuType",
    "AnyProvisioning",
    "DockerContainer",
    "DockerPullPolicy",
    "NoContainer",
    "AutoPythonEnv",
    "ManualPythonEnv",
    "maximum_score_function",
    "minimum_score_function",
    "score",
    "File",
    "whiteboard_",
    "whiteboard",
    "is_lzy_proxy",
    "materialize",
    "materialized",
    "__version__",
]
