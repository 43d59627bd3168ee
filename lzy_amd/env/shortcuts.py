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


# ---------------------------------------------------------------------------
# Reference-named shortcut spellings (pylzy/lzy/env/shortcuts.py:28-103).
# The reference vends appliers that mutate a decorated subject; here every
# shortcut returns an LzyEnvironment for `@op(env=...)` / `.with_env(...)`
# (see docs/migrating.md) — same names, same keyword signatures.
# ---------------------------------------------------------------------------


def docker_container(*, registry: str, image: str,
                     pull_policy=None, username: Optional[str] = None,
                     password: Optional[str] = None) -> LzyEnvironment:
    from lzy_amd.env.compat import DockerContainer, DockerPullPolicy

    DockerContainer(
        registry=registry, image=image,
        pull_policy=pull_policy or DockerPullPolicy.IF_NOT_EXISTS,
        username=username, password=password,
    ).validate()
    return LzyEnvironment()


def no_container() -> LzyEnvironment:
    """In-process execution IS the no-container mode."""
    return LzyEnvironment()


def manual_python_env(*, python_version: str,
                      local_module_paths: Optional[Sequence[str]] = None,
                      pypi_packages: Optional[dict] = None,
                      pypi_index_url: Optional[str] = None) -> LzyEnvironment:
    from lzy_amd.env.compat import ManualPythonEnv

    ManualPythonEnv(
        python_version=python_version,
        local_module_paths=tuple(local_module_paths or ()),
        pypi_packages=dict(pypi_packages or {}),
        pypi_index_url=pypi_index_url,
    ).validate()
    return LzyEnvironment()


def auto_python_env(*, pypi_index_url: Optional[str] = None,
                    additional_pypi_packages: Optional[dict] = None) -> LzyEnvironment:
    from lzy_amd.env.compat import AutoPythonEnv

    AutoPythonEnv(
        pypi_index_url=pypi_index_url,
        additional_pypi_packages=additional_pypi_packages,
    ).validate()
    return LzyEnvironment()


def provisioning(*, cpu_type=None, cpu_count=None, gpu_type=None,
                 gpu_count=None, ram_size_gb=None) -> LzyEnvironment:
    """Reference keyword spelling (`Any` sentinel accepted for every
    field; cpu_type has no meaning on a homogeneous node)."""
    from lzy_amd.env.provisioning import Provisioning, _norm

    if isinstance(gpu_type, object) and gpu_type is not None:
        gpu_type = _norm(gpu_type)
        if hasattr(gpu_type, "value"):  # GpuType enum member
            gpu_type = gpu_type.value
    return LzyEnvironment(provisioning=Provisioning(
        cpu_count=_norm(cpu_count),
        ram_size_gb=_norm(ram_size_gb),
        gpu_count=_norm(gpu_count),
        gpu_type=gpu_type,
    ))
