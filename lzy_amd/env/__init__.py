from lzy_amd.env.environment import LzyEnvironment
from lzy_amd.env.provisioning import GpuType, Provisioning

__all__ = ["LzyEnvironment", "Provisioning", "GpuType"]
