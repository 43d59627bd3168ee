"""Lzy entry object (reference: pylzy/lzy/core/lzy.py:46-176).

Differences by design: no cloud auth (the "cluster" is this node's 8
GPUs), the default runtime is picked from the launch context (torchrun ->
GpuPoolRuntime, plain python -> LocalRuntime), and the whiteboard index
is a sqlite DB beside the storage root instead of a remote service.
"""
from __future__ import annotations

import datetime
import inspect
import os
import sys
from dataclasses import dataclass, field
from functools import cached_property
from typing import Any, Iterable, Optional, Sequence

from lzy_amd.core.workflow import LzyWorkflow
from lzy_amd.env.environment import LzyEnvironment, WithEnvironmentMixin
from lzy_amd.runtime.base import Runtime
from lzy_amd.serialization.registry import LzySerializerRegistry
from lzy_amd.storage.api import StorageConfig, StorageRegistry
from lzy_amd.storage.fs import uri_to_path
from lzy_amd.whiteboards.index import WhiteboardIndexClient
from lzy_amd.whiteboards.wb import ReadOnlyWhiteboard


def lzy_auth(*, user: str, key_path: str = "", endpoint: str = "",
             whiteboards_endpoint: str = "") -> None:
    """Source-compatibility shim for the reference's cloud auth
    (reference: pylzy/lzy/core/lzy.py:27-43).  The single-node runtime has
    no remote service to authenticate against; the user name is recorded
    for log/metadata attribution."""
    os.environ["LZY_USER"] = user
    if key_path:
        os.environ["LZY_KEY_PATH"] = key_path


def _default_runtime() -> Runtime:
    # under torchrun (one process per GPU) the pool runtime is the engine;
    # otherwise the in-process local runtime.  Inside an op body (pool
    # worker thread) a nested Lzy() must NOT try to become a pool driver:
    # nested workflows run in-process (scenarios/nested_workflows).
    from lzy_amd.runtime.context import in_op_execution

    if (
        not in_op_execution()
        and not os.environ.get("LZY_INSIDE_OP")  # op-spawned subprocess
        and os.environ.get("WORLD_SIZE")
        and int(os.environ.get("WORLD_SIZE", "1")) > 1
    ):
        from lzy_amd.runtime.pool import GpuPoolRuntime

        return GpuPoolRuntime()
    from lzy_amd.runtime.local import LocalRuntime

    return LocalRuntime()


@dataclass(frozen=True)
class Lzy(WithEnvironmentMixin):
    env: LzyEnvironment = field(default_factory=LzyEnvironment)
    runtime: Runtime = field(default_factory=_default_runtime)
    storage_registry: StorageRegistry = field(default_factory=StorageRegistry)
    serializer_registry: LzySerializerRegistry = field(default_factory=LzySerializerRegistry)

    def __post_init__(self):
        if self.storage_registry.default_client() is None:
            cfg = self.runtime.storage()
            if cfg is not None:
                self.storage_registry.register_storage(
                    "provided_default_storage", cfg, default=True
                )
        self._maybe_serve_endpoints()

    _ENDPOINTS_SERVED = {"metrics": False, "status": False}

    def _maybe_serve_endpoints(self) -> None:
        """Honor config metrics_port/status_port: serve Prometheus text
        and the dashboard when configured (reference: Prometheus reporter
        per service + the web UI; opt-in here, once per process)."""
        from lzy_amd.config import get_config

        cfg = get_config()
        served = Lzy._ENDPOINTS_SERVED
        if cfg.metrics_port > 0 and not served["metrics"]:
            from lzy_amd.utils.metrics import METRICS

            METRICS.serve(cfg.metrics_port)
            served["metrics"] = True
        if cfg.status_port > 0 and not served["status"]:
            from lzy_amd.storage.fs import uri_to_path
            from lzy_amd.utils.status import serve_status

            root = None
            c = self.storage_registry.default_config()
            if c is not None and c.uri.startswith("file://"):
                root = str(uri_to_path(c.uri))
            serve_status(storage_root=root, port=cfg.status_port)
            served["status"] = True

    def auth(self, *, user: str, key_path: str = "", endpoint: str = "",
             whiteboards_endpoint: str = "") -> "Lzy":
        lzy_auth(user=user, key_path=key_path, endpoint=endpoint,
                 whiteboards_endpoint=whiteboards_endpoint)
        return self

    # -- storage ------------------------------------------------------------

    @cached_property
    def storage_name(self) -> str:
        name = self.storage_registry.default_storage_name()
        if name is None:
            raise ValueError("Default storage is not configured")
        return name

    @cached_property
    def storage_uri(self) -> str:
        cfg = self.storage_registry.default_config()
        if cfg is None:
            raise ValueError("Default storage is not configured")
        return cfg.uri

    @cached_property
    def storage_client(self):
        client = self.storage_registry.default_client()
        if client is None:
            raise ValueError("Default storage is not configured")
        return client

    @cached_property
    def whiteboard_index(self) -> WhiteboardIndexClient:
        root = uri_to_path(self.storage_uri)
        root.mkdir(parents=True, exist_ok=True)
        return WhiteboardIndexClient(str(root / "whiteboards.db"))

    @property
    def whiteboard_manager(self) -> WhiteboardIndexClient:
        """Reference-named alias (pylzy/lzy/core/lzy.py:85
        ``whiteboard_manager``) for the whiteboard index."""
        return self.whiteboard_index

    # -- workflows ----------------------------------------------------------

    def workflow(
        self,
        name: str,
        *,
        eager: bool = False,
        interactive: bool = True,
        env: Optional[LzyEnvironment] = None,
    ) -> LzyWorkflow:
        # sys._getframe(1) is the caller frame without inspect.stack()'s
        # full walk + source loading (findsource per frame — milliseconds)
        frame = sys._getframe(1)
        namespace = {**frame.f_globals, **frame.f_locals}
        wf_env = env or LzyEnvironment()
        wf_env = self.env.combine(
            wf_env.with_fields(namespace={**namespace, **wf_env.get_namespace()})
        )
        return LzyWorkflow(
            name=name, owner=self, env=wf_env, eager=eager, interactive=interactive
        )

    # -- whiteboard reads ---------------------------------------------------

    def whiteboard(self, id_: Optional[str] = None, **kw) -> Optional[Any]:
        if id_ is None:
            id_ = kw.get("id_")
        if id_ is None:
            raise ValueError("id_ is required")
        meta = self.whiteboard_index.get(id_)
        if meta is None:
            return None
        return ReadOnlyWhiteboard(meta, self)

    def executions(self, limit: int = 100) -> Sequence[dict]:
        """Recent executions with task-state counts from the crash-resume
        journals (reference: the site/frontend task listing; served over
        HTTP by utils/status.py — this is the in-process accessor)."""
        import tempfile

        from lzy_amd.utils.status import _workflows_payload

        jdir = os.path.join(tempfile.gettempdir(), "lzy_amd_journal")
        return _workflows_payload(jdir)[:limit]

    def whiteboards(
        self,
        *,
        name: Optional[str] = None,
        tags: Sequence[str] = (),
        not_before: Optional[datetime.datetime] = None,
        not_after: Optional[datetime.datetime] = None,
    ) -> Iterable[Any]:
        for meta in self.whiteboard_index.query(
            name=name, tags=tags, not_before=not_before, not_after=not_after
        ):
            yield ReadOnlyWhiteboard(meta, self)
