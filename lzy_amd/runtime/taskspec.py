"""Picklable task description + worker-side execution.

Re-design of the reference's worker workload: where the reference embeds a
pickled ProcessingRequest in a command line and re-reads inputs from slot
files (reference: pylzy api/v1/remote/runtime.py:368-384 +
api/v1/startup.py:109-185), the pool sends a compact TaskSpec over the
control plane and the worker reads inputs straight from its in-process
store (already landed there by RCCL/xGMI transfers).
"""
from __future__ import annotations

import logging
import os
import time
import traceback
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

from lzy_amd.runtime.executor import cache_base_uri
from lzy_amd.utils.logs import OpLogCapture
from lzy_amd.utils.metrics import METRICS

_LOG = logging.getLogger("lzy_amd.taskspec")


@dataclass
class TaskSpec:
    task_id: str
    name: str
    func_bytes: bytes
    arg_entries: List[str]
    kwarg_entries: Dict[str, str]
    output_entries: List[Tuple[str, str]]  # (entry_id, storage_uri)
    exception_entry: str
    env_vars: Dict[str, str] = field(default_factory=dict)
    execution_id: str = ""
    cache: bool = False
    version: str = "0.0"
    storage_root: str = ""
    inline_values: Dict[str, bytes] = field(default_factory=dict)
    wait_entries: List[str] = field(default_factory=list)
    gang: Optional[dict] = None  # {"ranks": [...], "gang_rank": i, "tag": str}


@dataclass
class TaskResult:
    task_id: str
    ok: bool
    outputs: List[dict] = field(default_factory=list)  # EntryMeta wire dicts
    exc_bytes: Optional[bytes] = None
    elapsed_s: float = 0.0
    cached: bool = False
    # captured op std-logs, tailed on the driver's console (reference:
    # worker -> Kafka -> client ReadStdSlots stream); capped at 256 KiB
    logs_out: str = ""
    logs_err: str = ""


class _PoisonEntry:
    """Marks an entry whose producer failed: chained consumers waiting on
    it must fail promptly instead of timing out."""

    def __init__(self, reason: str):
        self.reason = reason


class WorkerStore:
    """Per-rank value store: entry_id -> live python object (device tensors
    stay in HBM).  Non-tensor values keep a pre-pickled byte image so
    transfer sizes are known up front (single-phase recv).

    Past the HBM threshold, LRU device tensors spill to pinned host
    memory and come back transparently at get() — the 288 GB-per-GPU
    result store with pinned-host async spill (storage/spill.py)."""

    def __init__(self, device=None) -> None:
        import threading

        self.values: Dict[str, Any] = {}
        self.pickled: Dict[str, bytes] = {}
        self._cond = threading.Condition()
        from lzy_amd.storage.spill import SpillManager

        self.spill = SpillManager(device=device)
        # config resolved once: wait_present runs on the per-task settle
        # path, and get_config() re-fingerprints the environment
        from lzy_amd.config import get_config

        self._settle_wait = float(getattr(get_config(), "settle_wait_s", 120.0))

    def put(self, entry_id: str, value: Any, pickled: Optional[bytes] = None) -> None:
        from lzy_amd.runtime.streams import STREAMS

        # the producing event must exist BEFORE publication: a chained
        # consumer wakes on the condition notify and immediately waits on
        # the event.  Producers that already recorded a precise
        # (op-stream) event win — if_absent keeps theirs.
        STREAMS.record_output(entry_id, value, if_absent=True)
        with self._cond:
            self.values[entry_id] = value
            if pickled is not None:
                self.pickled[entry_id] = pickled
            self._cond.notify_all()
        self.spill.track(entry_id, value)
        self.spill.maybe_spill(self.values)

    def wait_present(self, entry_id: str, timeout: Optional[float] = None) -> bool:
        """Block until the entry lands in the store (another task's settle
        may be completing the transfer concurrently).  Default timeout is
        config-derived (settle_wait_s, resolved at store construction)."""
        if timeout is None:
            timeout = self._settle_wait
        with self._cond:
            return self._cond.wait_for(
                lambda: entry_id in self.values, timeout=timeout
            )

    def poison(self, entry_id: str, reason: str) -> None:
        """Producer failed: wake and fail any chained consumer waiting on
        this entry."""
        with self._cond:
            if entry_id not in self.values:
                self.values[entry_id] = _PoisonEntry(reason)
                self._cond.notify_all()

    def get(self, entry_id: str) -> Any:
        # unspill returns None when the entry is not spilled — including
        # when a concurrent get() unspilled it first, in which case the
        # device tensor is already back in self.values (written under the
        # spill lock): fall through instead of returning None.
        value = self.spill.unspill(entry_id, self.values)
        if value is None:
            value = self.values[entry_id]
            self.spill.track(entry_id, value)  # LRU touch
        if isinstance(value, _PoisonEntry):
            raise RuntimeError(f"input {entry_id} unavailable: {value.reason}")
        from lzy_amd.runtime.streams import STREAMS

        STREAMS.wait_value(entry_id, value)  # order after producing stream
        return value

    def has(self, entry_id: str) -> bool:
        return entry_id in self.values

    def drop(self, entry_id: str) -> None:
        self.values.pop(entry_id, None)
        self.pickled.pop(entry_id, None)
        self.spill.forget(entry_id)
        from lzy_amd.runtime.streams import STREAMS

        STREAMS.drop(entry_id)

    def clear(self) -> None:
        self.values.clear()
        self.pickled.clear()


_LOG_CAP = 256 << 10
_INLINE_RESULT_LIMIT = 64 << 10  # small results piggyback on task_done


def run_taskspec(
    spec: TaskSpec,
    store: WorkerStore,
    serializers,
    storage,
    gang_group=None,
    echo_logs: bool = True,
    live_sink=None,
) -> TaskResult:
    """Execute one TaskSpec against the worker store (worker innermost loop)."""
    from lzy_amd.channels.transport import describe_value, pickle_value, unpickle_value
    from lzy_amd.snapshot import hash_value

    import torch

    t0 = time.perf_counter()

    for eid, data in spec.inline_values.items():
        if not store.has(eid):
            store.put(eid, unpickle_value(data), pickled=data)

    # -- result cache (worker-side CheckCache; rank-agnostic key) -----------
    out_uris = dict(spec.output_entries)
    if spec.cache:
        hashes = [
            hash_value(store.get(eid), serializers)
            for eid in list(spec.arg_entries) + list(spec.kwarg_entries.values())
        ]
        base = cache_base_uri(spec.storage_root, spec.name, spec.version, hashes)
        out_uris = {
            eid: f"{base}/return_{i}"
            for i, (eid, _) in enumerate(spec.output_entries)
        }
        if all(storage.blob_exists(u) for u in out_uris.values()):
            outputs = []
            for eid, uri in out_uris.items():
                data = storage.read_bytes(uri)
                fmt = _read_fmt(storage, uri)  # format recorded beside the blob
                value = serializers.loads(data, fmt)
                store.put(eid, value)
                meta = describe_value(eid, value)
                outputs.append({**meta.to_wire(), "uri": uri})
            METRICS.inc("lzy_cache_hits", op=spec.name)
            return TaskResult(
                task_id=spec.task_id, ok=True, outputs=outputs,
                elapsed_s=time.perf_counter() - t0, cached=True,
            )

    # -- materialize inputs --------------------------------------------------
    try:
        args = [store.get(eid) for eid in spec.arg_entries]
        kwargs = {k: store.get(eid) for k, eid in spec.kwarg_entries.items()}
    except KeyError as e:
        return _fail(spec, t0, RuntimeError(f"missing input entry {e}"), "")

    func = _load_func(spec.func_bytes)

    capture = OpLogCapture.instance()
    out_buf, err_buf = capture.route_current_thread(spec.name, echo=echo_logs)
    # live log streaming (reference: worker→Kafka→client ReadStdSlots
    # stream while the op RUNS, KafkaLogsListeners.java:35): the agent's
    # flusher ships buffer deltas to the driver until unregistered
    unregister_live = live_sink(out_buf, err_buf) if live_sink else None
    old_env: Dict[str, Optional[str]] = {}
    # subprocesses spawned BY the op (user DDP launchers etc.) inherit
    # this marker and must not try to become pool drivers (reference:
    # the LZY_OP_MAIN_PID guard, startup.py:80-106)
    gang_env: Dict[str, str] = {"LZY_INSIDE_OP": "1"}
    if spec.gang is not None:
        gang_env.update({
            "LZY_OP_RANK": str(spec.gang["gang_rank"]),
            "LZY_OP_WORLD_SIZE": str(len(spec.gang["ranks"])),
        })
    try:
        for k, v in {**spec.env_vars, **gang_env}.items():
            old_env[k] = os.environ.get(k)
            os.environ[k] = v
        if spec.gang is not None:
            from lzy_amd.runtime.context import _set_op_context, OpContext

            _set_op_context(OpContext(
                gang_rank=spec.gang["gang_rank"],
                gang_size=len(spec.gang["ranks"]),
                ranks=tuple(spec.gang["ranks"]),
                process_group=gang_group,
            ))
        # per-task HIP stream: back-to-back tasks on this rank overlap on
        # device; input hand-off is event-ordered (runtime/streams.py)
        from contextlib import nullcontext

        from lzy_amd.runtime.streams import STREAMS

        from lzy_amd.runtime.context import _set_in_op_execution

        stream = STREAMS.next_stream()
        _set_in_op_execution(True)
        with (torch.cuda.stream(stream) if stream is not None else nullcontext()):
            for eid, v in zip(
                list(spec.arg_entries) + list(spec.kwarg_entries.values()),
                args + list(kwargs.values()),
            ):
                STREAMS.wait_value(eid, v)
            result = func(*args, **kwargs)
    except BaseException as e:  # noqa: BLE001 - transported as a value
        tr = _fail(spec, t0, e, traceback.format_exc())
        tr.logs_out = out_buf.getvalue()[:_LOG_CAP]
        tr.logs_err = err_buf.getvalue()[:_LOG_CAP]
        return tr
    finally:
        if unregister_live is not None:
            unregister_live()  # final flush precedes the completion event
        _set_in_op_execution(False)
        if spec.gang is not None:
            from lzy_amd.runtime.context import _set_op_context

            _set_op_context(None)
        for k, v in old_env.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v
        capture.unroute_current_thread()
        _archive_logs(spec, storage, out_buf.getvalue(), err_buf.getvalue())

    # -- store outputs -------------------------------------------------------
    n = len(spec.output_entries)
    if n == 1:
        outs: Tuple[Any, ...] = (result,)
    elif isinstance(result, tuple) and len(result) == n:
        outs = result
    else:
        return _fail(
            spec, t0,
            RuntimeError(
                f"op {spec.name} declared {n} outputs, returned {type(result).__name__}"
            ),
            "",
        )

    outputs = []
    gang_primary = spec.gang is None or spec.gang["gang_rank"] == 0
    for (eid, _), value in zip(spec.output_entries, outs):
        pickled = None
        if not isinstance(value, torch.Tensor):
            pickled = pickle_value(value)
        from lzy_amd.runtime.streams import STREAMS as _S

        _S.record_output(eid, value, stream=stream)  # precise, pre-publication
        store.put(eid, value, pickled=pickled)
        meta = describe_value(eid, value)
        wire = meta.to_wire()
        if pickled is not None:
            meta.nbytes = wire["nbytes"] = len(pickled)
            if len(pickled) <= _INLINE_RESULT_LIMIT:
                # small host results ride the completion event: the
                # driver can hand them out without a fetch round trip
                wire["inline"] = pickled
        outputs.append(wire)
        if spec.cache and gang_primary:
            _S.wait_value(eid, value)  # D2H serialize orders after the op
            data, fmt = serializers.dumps(value)
            # sidecar BEFORE the data blob: cache hits gate on the data
            # blob's existence, so the format must already be readable
            _write_fmt(storage, out_uris[eid], fmt)
            storage.write_bytes(out_uris[eid], data)
            # the driver must learn the durable location on the WRITE
            # path too (not only on later cache hits): failover re-roots
            # lost inputs from this blob (the channel's storage peer)
            wire["uri"] = out_uris[eid]

    METRICS.inc("lzy_op_runs", op=spec.name)
    elapsed = time.perf_counter() - t0
    METRICS.observe("lzy_op_run", elapsed)
    METRICS.observe(f"lzy_op::{spec.name}", elapsed)
    return TaskResult(
        task_id=spec.task_id, ok=True, outputs=outputs, elapsed_s=elapsed,
        logs_out=out_buf.getvalue()[:_LOG_CAP],
        logs_err=err_buf.getvalue()[:_LOG_CAP],
    )


def _archive_logs(spec: TaskSpec, storage, out: str, err: str) -> None:
    """Worker-side std-log archive (reference s3-sink analogue)."""
    if (not out and not err) or not spec.storage_root:
        return
    try:
        base = (
            f"{spec.storage_root}/lzy_logs/{spec.execution_id}"
            f"/{spec.name}-{spec.task_id[:8]}"
        )
        if out:
            storage.write_bytes(base + ".out", out.encode())
        if err:
            storage.write_bytes(base + ".err", err.encode())
    except Exception:  # noqa: BLE001
        _LOG.warning("failed to archive logs for %s", spec.name)


_FUNC_CACHE: Dict[int, Any] = {}


def _load_func(func_bytes: bytes):
    """Worker-side unpickle memo keyed by content hash of the bytes."""
    from lzy_amd.channels.transport import unpickle_value
    from lzy_amd.sched import xxhash64

    key = xxhash64(func_bytes)
    func = _FUNC_CACHE.get(key)
    if func is None:
        func = unpickle_value(func_bytes)
        _FUNC_CACHE[key] = func
    return func


def _fail(spec: TaskSpec, t0: float, exc: BaseException, tb: str) -> TaskResult:
    from lzy_amd.channels.transport import pickle_value

    METRICS.inc("lzy_op_failures", op=spec.name)
    # ship the exception OBJECT too (reference: pickled exc_info written
    # as a first-class output, startup.py:124-169) so the client re-raises
    # the user's exception type with its attributes.  Packed as
    # (class, args, __dict__) and rebuilt with __new__ — default pickling
    # breaks on exception classes whose __init__ signature differs from
    # their .args.  Falls back to the string triple when unpicklable.
    exc_blob: Optional[bytes] = None
    try:
        exc_blob = pickle_value(
            (type(exc), tuple(exc.args), dict(getattr(exc, "__dict__", {})))
        )
    except Exception:  # noqa: BLE001
        pass
    payload = pickle_value((type(exc).__name__, str(exc), tb, exc_blob))
    return TaskResult(
        task_id=spec.task_id, ok=False, exc_bytes=payload,
        elapsed_s=time.perf_counter() - t0,
    )


def _write_fmt(storage, uri: str, fmt: str) -> None:
    storage.write_bytes(uri + ".fmt", fmt.encode())


def _read_fmt(storage, uri: str) -> str:
    try:
        return storage.read_bytes(uri + ".fmt").decode()
    except Exception:
        return "pickle"
