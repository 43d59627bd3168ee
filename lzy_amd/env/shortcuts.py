"""Env shortcut helpers (reference: pylzy/lzy/env/shortcuts.py).

The reference vends fluent helpers building python-env / container /
provisioning pieces.  On one node the python env IS the driver's env and
there are no containers to pull, so the python/container shortcuts are
accepted-and-validated no-ops that keep reference user scripts running
unchanged; provisioning shortcuts are fully functional.
"""
from __future__ import annotations

import sys
import warnings
from typing import Optional, Sequence

from lzy_amd.env.environment import LzyEnvironment
from lzy_amd.env.provisioning import Provisioning


def gpu(count: int = 1, type_: str = "MI355X") -> LzyEnvironment:
    return LzyEnvironment(provisioning=Provisioning(gpu_count=count, gpu_type=type_))


def cpu(count: int) -> LzyEnvironment:
    return LzyEnvironment(provisioning=Provisioning(cpu_count=count))


def ram(size_gb: int) -> LzyEnvironment:
    return LzyEnvironment(provisioning=Provisioning(ram_size_gb=size_gb))


def env_vars(**variables: str) -> LzyEnvironment:
    return LzyEnvironment(env_variables=dict(variables))


def manual_python(python_version: Optional[str] = None,
                  libraries: Optional[dict] = None) -> LzyEnvironment:
    """Reference: pin python version + libs for the remote conda env.

    Single-node semantics: workers share this interpreter, so instead of
    INSTALLING the request (CondaEnvironment.java:67-134) we VALIDATE
    it — a library that is not importable here cannot be provisioned and
    fails fast like an unsatisfiable pool; a version mismatch warns (the
    env exists but differs from the pin)."""
    if python_version is not None:
        current = f"{sys.version_info.major}.{sys.version_info.minor}"
        if not python_version.startswith(current):
            warnings.warn(
                f"manual_python({python_version!r}) ignored: workers run "
                f"in-process on python {current} (no env re-provisioning "
                f"on a single node)"
            )
    for name, want in (libraries or {}).items():
        try:
            from importlib import metadata

            have = metadata.version(name)
        except Exception:
            from lzy_amd.exceptions import BadProvisioningError

            raise BadProvisioningError(
                f"manual_python requires library {name!r}, which is not "
                "installed in this image — a single-node pool cannot "
                "install packages (no conda re-provisioning by design)"
            )
        if want and str(want) not in (have, ""):
            if not str(have).startswith(str(want)):
                warnings.warn(
                    f"manual_python: {name} pinned to {want} but the "
                    f"shared interpreter has {have}"
                )
    return LzyEnvironment()


def auto_python() -> LzyEnvironment:
    """Reference: explore imports and sync them to the worker.  Workers
    import the same site-packages — nothing to sync."""
    return LzyEnvironment()


def docker_image(image: str, pull_policy: str = "IF_NOT_EXISTS") -> LzyEnvironment:
    """Reference: run the op inside a docker container.  Not applicable to
    the in-process pool; warns and proceeds without a container."""
    warnings.warn(
        f"docker_image({image!r}) ignored: ops run in-process on this node "
        f"(no container runtime in the data plane)"
    )
    return LzyEnvironment()
