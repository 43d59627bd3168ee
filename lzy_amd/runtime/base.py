"""Runtime interface.

Reference capability (pylzy/lzy/api/v1/runtime.py): start / exec / abort /
finish / storage vending.  Two implementations here, matching the
reference's RemoteRuntime/LocalRuntime split re-thought for one MI355X
node:

  * ``LocalRuntime``   — in-process scheduler + thread pool (CPU ops and
    single-GPU ops on the driver's GPU);
  * ``GpuPoolRuntime`` — one worker process per GPU over torch.distributed
    (RCCL for tensors, gloo for control), the flagship.
"""
from __future__ import annotations

import abc
from typing import TYPE_CHECKING, List, Optional, Sequence

from lzy_amd.storage.api import StorageConfig

if TYPE_CHECKING:
    from lzy_amd.core.call import LzyCall
    from lzy_amd.core.workflow import LzyWorkflow


class Runtime(abc.ABC):
    @abc.abstractmethod
    def storage(self) -> Optional[StorageConfig]:
        """Default storage this runtime vends (reference GetOrCreateDefaultStorage)."""

    @abc.abstractmethod
    def start(self, workflow: "LzyWorkflow") -> None: ...

    @abc.abstractmethod
    def exec(self, workflow: "LzyWorkflow", calls: Sequence["LzyCall"]) -> None:
        """Execute one graph (a barrier batch); raises on op failure."""

    @abc.abstractmethod
    def finish(self, workflow: "LzyWorkflow") -> None: ...

    @abc.abstractmethod
    def abort(self, workflow: "LzyWorkflow") -> None: ...
