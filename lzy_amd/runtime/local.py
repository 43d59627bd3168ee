"""LocalRuntime: in-process DAG execution on the driver node.

Reference capability (pylzy/lzy/api/v1/local/runtime.py:30-201): run the
captured DAG without a cluster.  Re-design: instead of topo-sort +
sequential subprocesses with slot files, a ready-frontier scheduler
(lzy_amd.sched, C++ when built) drives a thread pool; ops share the
process so data passes by reference (device tensors never move).  Ops
declaring ``gpu_count>1`` are rejected here — that's GpuPoolRuntime's job;
``gpu_count=1`` ops run on the driver's default GPU when present.

Dispatch overhead target: sub-millisecond per op (vs the reference's 1 s
scheduler tick + 10 s status poll — BASELINE.md).
"""
from __future__ import annotations

import logging
import os
import tempfile
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from typing import TYPE_CHECKING, Dict, List, Optional, Sequence

from lzy_amd.exceptions import BadProvisioningError, WorkflowAbortedError
from lzy_amd.runtime.base import Runtime
from lzy_amd.runtime.executor import execute_call
from lzy_amd.sched import Dag, Journal
from lzy_amd.storage.api import StorageConfig
from lzy_amd.utils.faults import FAULTS
from lzy_amd.utils.logs import OpLogCapture
from lzy_amd.utils.metrics import METRICS

if TYPE_CHECKING:
    from lzy_amd.core.call import LzyCall
    from lzy_amd.core.workflow import LzyWorkflow

_LOG = logging.getLogger("lzy_amd.local_runtime")

DEFAULT_STORAGE_ENV = "LZY_AMD_STORAGE"


def default_storage_uri() -> str:
    from lzy_amd.config import get_config

    root = get_config().storage or os.path.join(
        tempfile.gettempdir(), "lzy_amd_storage"
    )
    if root.startswith("file://"):
        return root
    return f"file://{root}"


class LocalRuntime(Runtime):
    def __init__(self, max_workers: Optional[int] = None, journal_dir: Optional[str] = None):
        if max_workers is None:
            from lzy_amd.config import get_config

            max_workers = get_config().dispatch_workers or None
        self._max_workers = max_workers or min(32, (os.cpu_count() or 8))
        self._journal_dir = journal_dir
        self._pool: Optional[ThreadPoolExecutor] = None
        self._journal: Optional[Journal] = None
        # single-flight: one workflow at a time per runtime instance —
        # concurrent client threads serialize here instead of racing the
        # journal/pool state (the reference serializes per execution via
        # its DB; nested workflows use their own Lzy()/runtime)
        import threading as _threading

        # opportunistic journal GC (same sweep as GpuPoolRuntime)
        try:
            from lzy_amd.config import get_config
            from lzy_amd.storage.gc import gc_journals

            ttl_h = float(getattr(get_config(), "journal_ttl_hours", 168.0))
            if ttl_h > 0:
                gc_journals(
                    self._journal_dir or os.path.join(
                        tempfile.gettempdir(), "lzy_amd_journal"
                    ),
                    ttl_seconds=ttl_h * 3600.0,
                )
        except Exception:  # noqa: BLE001 - GC must never block startup
            pass
        self._flight = _threading.Lock()
        # Lock has no ownership: remember WHICH workflow holds the flight
        # so finish/abort of a never-started (or already-finished) one
        # cannot release another thread's active flight.
        self._flight_owner: Optional[str] = None

    def storage(self) -> Optional[StorageConfig]:
        return StorageConfig(uri=default_storage_uri())

    def start(self, workflow: "LzyWorkflow") -> None:
        self._flight.acquire()
        self._flight_owner = workflow.execution_id
        self._pool = ThreadPoolExecutor(
            max_workers=self._max_workers, thread_name_prefix="lzy-op"
        )
        jdir = self._journal_dir or os.path.join(
            tempfile.gettempdir(), "lzy_amd_journal"
        )
        self._journal = Journal(os.path.join(jdir, f"{workflow.execution_id}.jsonl"))
        OpLogCapture.instance().install()

    def exec(self, workflow: "LzyWorkflow", calls: Sequence["LzyCall"]) -> None:
        FAULTS.hit("runtime.exec.begin")
        t_build0 = time.perf_counter()
        by_id: Dict[str, "LzyCall"] = {c.id: c for c in calls}
        producer: Dict[str, str] = {}
        for c in calls:
            for eid in c.entry_ids:
                producer[eid] = c.id
            producer[c.exception_id] = c.id

        self._validate_provisioning(calls)

        dag = Dag()
        for c in calls:
            deps = sorted(
                {
                    producer[eid]
                    for eid in c.input_entry_ids()
                    if eid in producer and producer[eid] != c.id
                }
            )
            dag.add_task(c.id, deps)
        dag.seal()
        METRICS.observe("lzy_graph_build", time.perf_counter() - t_build0)

        lock = threading.Lock()
        done_cv = threading.Condition(lock)
        errors: List[BaseException] = []
        inflight = {"n": 0}

        def submit(tid: str) -> None:
            inflight["n"] += 1
            self._journal.record(tid, "scheduled", by_id[tid].callable_name)
            dispatch_t0 = time.perf_counter()
            self._pool.submit(run, tid, dispatch_t0)

        def run(tid: str, dispatch_t0: float) -> None:
            METRICS.observe("lzy_dispatch", time.perf_counter() - dispatch_t0)
            call = by_id[tid]
            try:
                self._journal.record(tid, "running")
                execute_call(call)
                with lock:
                    self._journal.record(tid, "done")
                    newly = dag.complete(tid)
                    inflight["n"] -= 1
                    for nt in newly:
                        submit(nt)
                    done_cv.notify_all()
            except BaseException as e:  # noqa: BLE001
                with lock:
                    self._journal.record(tid, "failed", str(e))
                    cancelled = dag.fail(tid)
                    for ct in cancelled:
                        self._journal.record(ct, "cancelled")
                    errors.append(e)
                    inflight["n"] -= 1
                    done_cv.notify_all()

        with lock:
            for tid in dag.take_ready():
                submit(tid)
            while inflight["n"] > 0:
                done_cv.wait()

        if errors:
            raise errors[0]

    def _validate_provisioning(self, calls: Sequence["LzyCall"]) -> None:
        import torch

        n_gpus = torch.cuda.device_count() if torch.cuda.is_available() else 0
        for c in calls:
            need = c.env.provisioning.effective_gpu_count
            if need > max(n_gpus, 0) and need > 0:
                raise BadProvisioningError(
                    f"Op {c.callable_name} requests gpu_count={need} but "
                    f"LocalRuntime sees {n_gpus} GPUs; use GpuPoolRuntime "
                    f"(torchrun) for multi-GPU ops"
                )

    def finish(self, workflow: "LzyWorkflow") -> None:
        self._teardown(workflow)

    def abort(self, workflow: "LzyWorkflow") -> None:
        self._teardown(workflow)

    def _teardown(self, workflow: "LzyWorkflow") -> None:
        if self._flight_owner != workflow.execution_id:
            return  # this workflow never started (or already finished)
        torn = self._pool is not None
        if self._pool is not None:
            self._pool.shutdown(wait=True)
            self._pool = None
        if self._journal is not None:
            self._journal.close()
            self._journal = None
        self._flight_owner = None
        try:
            self._flight.release()
        except RuntimeError:
            pass
        if torn:  # balance exactly one uninstall per start
            OpLogCapture.instance().uninstall()
