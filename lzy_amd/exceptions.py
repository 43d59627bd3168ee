"""Framework exceptions.

Mirrors the error surface of the reference SDK (pylzy/lzy/api/v1/exceptions.py)
without copying it: a remote op failure is transported as a first-class
exception entry and re-raised at the barrier.
"""
from __future__ import annotations


class LzyError(Exception):
    """Base class for all framework errors."""


class LzyExecutionError(LzyError):
    """An @op raised during execution; carries the remote traceback text."""

    def __init__(self, message: str, task_id: str = "", remote_traceback: str = ""):
        super().__init__(message)
        self.task_id = task_id
        self.remote_traceback = remote_traceback


class WorkflowAbortedError(LzyError):
    """Workflow was aborted (user abort or fatal scheduler error)."""


class SerializationError(LzyError):
    """No serializer available/compatible for a value."""


class BadProvisioningError(LzyError):
    """Requested provisioning (gpu_count etc.) cannot be satisfied by the node."""


class ChannelError(LzyError):
    """Device-channel transfer failed (both direct and spill path)."""


class NativeExtensionMissing(LzyError):
    """A HIP/C++ extension is required on this platform but not built.

    On a GPU box the HIP ops must never silently fall back to eager torch;
    this error makes the missing extension loud.
    """
