"""Reference-named env classes for drop-in imports.

Reference capability (pylzy/lzy/api/v1/__init__.py:5-27 public exports;
env/container/docker.py:10-27, env/container/no_container.py:6,
env/python/manual.py:10-23, env/python/auto.py:24-36): user scripts
import DockerContainer / NoContainer / ManualPythonEnv / AutoPythonEnv
by name and attach them to ops.  On one MI355X node workers run
in-process in THIS interpreter, so the container classes validate and
warn (there is no container runtime in the data plane) and the python
envs VALIDATE the pin against the shared interpreter instead of
provisioning a conda env — same fail-fast semantics the allocator gives
an unsatisfiable pool, with zero provisioning latency.
"""
from __future__ import annotations

import enum
import os
import sys
import warnings
from dataclasses import dataclass, field
from typing import Dict, Optional, Sequence

from lzy_amd.exceptions import BadProvisioningError


class DockerPullPolicy(enum.Enum):
    ALWAYS = "ALWAYS"
    IF_NOT_EXISTS = "IF_NOT_EXISTS"


@dataclass(frozen=True)
class NoContainer:
    """The default 'run on the node itself' container — exactly what the
    in-process pool does, so this is the one container type that is
    fully native here."""

    def deconstruct(self) -> Dict[str, str]:
        return {}


@dataclass(frozen=True)
class DockerContainer:
    registry: str
    image: str
    pull_policy: DockerPullPolicy = DockerPullPolicy.IF_NOT_EXISTS
    username: Optional[str] = None
    password: Optional[str] = None

    def get_registry(self) -> str:
        return self.registry

    def get_image(self) -> str:
        return self.image

    def validate(self) -> None:
        """Ops run in-process; accepted with a warning so reference
        scripts keep running unchanged (the image is NOT pulled)."""
        warnings.warn(
            f"DockerContainer({self.registry}/{self.image}) ignored: ops "
            "run in-process on this node (no container runtime in the "
            "data plane)"
        )


@dataclass(frozen=True)
class ManualPythonEnv:
    """Pin python version / module paths / packages.

    Reference installs the pin into a fresh conda env per VM
    (CondaEnvironment.java:67-134); here the pin is VALIDATED against
    the shared interpreter: a missing package raises
    BadProvisioningError (like an unsatisfiable pool), a version
    mismatch warns, a missing module path warns."""

    python_version: str
    local_module_paths: Sequence[str] = ()
    pypi_packages: Dict[str, str] = field(default_factory=dict)
    pypi_index_url: Optional[str] = None

    def get_python_version(self) -> str:
        return self.python_version

    def validate(self) -> None:
        current = f"{sys.version_info.major}.{sys.version_info.minor}"
        if self.python_version and not self.python_version.startswith(current):
            warnings.warn(
                f"ManualPythonEnv({self.python_version!r}): workers run "
                f"in-process on python {current} (no env re-provisioning "
                "on a single node)"
            )
        for path in self.local_module_paths:
            if not os.path.exists(path):
                warnings.warn(
                    f"ManualPythonEnv: local module path {path!r} does not "
                    "exist on this node"
                )
        # reference flag (pylzy/lzy/config.py:34 LZY_SKIP_PYPI_VALIDATION):
        # opt out of package-pin validation entirely
        if os.environ.get("LZY_SKIP_PYPI_VALIDATION", "").lower() in (
                "1", "yes", "true", "on"):
            return
        from importlib import metadata

        for name, want in self.pypi_packages.items():
            try:
                have = metadata.version(name)
            except Exception:
                raise BadProvisioningError(
                    f"ManualPythonEnv requires package {name!r}, which is "
                    "not installed in this image — a single-node pool "
                    "cannot install packages (no conda re-provisioning by "
                    "design)"
                )
            if want and not str(have).startswith(str(want)):
                warnings.warn(
                    f"ManualPythonEnv: {name} pinned to {want} but the "
                    f"shared interpreter has {have}"
                )


@dataclass(frozen=True)
class AutoPythonEnv:
    """Reference explores imported modules and syncs them to the worker
    VM.  Workers import the same site-packages as the driver — the
    exploration result is the environment itself, so this is a fully
    satisfied no-op (extra pypi packages are validated if given)."""

    pypi_index_url: Optional[str] = None
    additional_pypi_packages: Optional[Dict[str, str]] = None

    def validate(self) -> None:
        if self.additional_pypi_packages:
            ManualPythonEnv(
                python_version="",
                pypi_packages=dict(self.additional_pypi_packages),
            ).validate()
