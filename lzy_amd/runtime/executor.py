"""In-process execution of one LzyCall.

This is the worker-side innermost loop — the re-design of the reference's
remote entrypoint (pylzy/lzy/api/v1/startup.py:109-185): read inputs,
run the op, write returns, transport exceptions as first-class outputs.
Differences by design: inputs are store references (no slot files, no
deserialization), std-logs are captured via a thread-routing tee, and the
op result cache is checked against the durable tier before running
(reference CheckCache.java:31-48 semantics, done worker-side).
"""
from __future__ import annotations

import logging
import os
import sys
import time
import traceback
from contextlib import contextmanager
from typing import TYPE_CHECKING, Any, Dict, Sequence, Tuple

from lzy_amd.exceptions import LzyExecutionError
from lzy_amd.proxy import is_lzy_proxy, lzy_proxy, materialize
from lzy_amd.utils.faults import FAULTS
from lzy_amd.utils.logs import OpLogCapture
from lzy_amd.utils.metrics import METRICS

if TYPE_CHECKING:
    from lzy_amd.core.call import LzyCall

_LOG = logging.getLogger("lzy_amd.executor")


@contextmanager
def _env_vars(env: Dict[str, str]):
    old: Dict[str, Any] = {}
    for k, v in env.items():
        old[k] = os.environ.get(k)
        os.environ[k] = v
    try:
        yield
    finally:
        for k, v in old.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v


def cache_base_uri(storage_uri: str, op_name: str, version: str, input_hashes) -> str:
    """Deterministic cache location: pure function of op name, version and
    input content hashes (reference workflow.py:247-281).  Shared by the
    local executor and the pool workers so cache hits are rank-agnostic."""
    import hashlib

    hasher = hashlib.blake2b(digest_size=16)
    hasher.update(op_name.encode())
    hasher.update(version.encode())
    for h in input_hashes:
        hasher.update(h.encode())
    return f"{storage_uri}/lzy_cache/{op_name}/{version}/{hasher.hexdigest()}"


def resolve_cache_uris(call: "LzyCall") -> None:
    """Point the call's output entries at deterministic cache URIs; runs at
    task start, when every input value is materialized."""
    snap = call.workflow.snapshot
    base = cache_base_uri(
        call.workflow.owner.storage_uri,
        call.callable_name,
        call.version,
        [snap.hash_of(eid) for eid in call.input_entry_ids()],
    )
    for i, eid in enumerate(call.entry_ids):
        snap.update_entry_uri(eid, f"{base}/return_{i}")


def cache_hit(call: "LzyCall") -> bool:
    """True when every cached output blob already exists (CheckCache parity)."""
    if not call.cache:
        return False
    snap = call.workflow.snapshot
    return all(
        snap.storage.blob_exists(snap.get_entry(eid).storage_uri)
        for eid in call.entry_ids
    )


def execute_call(call: "LzyCall") -> None:
    """Run one op in the current thread/process; raises LzyExecutionError."""
    snap = call.workflow.snapshot
    name = call.callable_name

    FAULTS.hit("executor.before_run")

    if call.cache:
        resolve_cache_uris(call)
    if cache_hit(call):
        METRICS.inc("lzy_cache_hits", op=name)
        for eid in call.entry_ids:
            snap.load(eid)
        return

    # materialize inputs (or hand over lazy proxies when lazy_arguments)
    args = []
    for eid, typ in zip(call.arg_entry_ids, call.signature.arg_types):
        args.append(_input_value(call, eid, typ))
    kwargs = {
        k: _input_value(call, eid, call.signature.kwarg_types.get(k, object))
        for k, eid in call.kwarg_entry_ids.items()
    }

    # per-op HIP stream: independent ops on one GPU overlap; inputs are
    # made safe with device-side event waits (lzy_amd/runtime/streams.py)
    from contextlib import nullcontext

    import torch

    from lzy_amd.runtime.streams import STREAMS

    stream = STREAMS.next_stream()
    stream_ctx = torch.cuda.stream(stream) if stream is not None else nullcontext()

    capture = OpLogCapture.instance()
    out_buf, err_buf = capture.route_current_thread(name)
    t0 = time.perf_counter()
    try:
        with _env_vars(call.env.env_variables), stream_ctx:
            for eid, v in zip(
                list(call.arg_entry_ids) + list(call.kwarg_entry_ids.values()),
                args + list(kwargs.values()),
            ):
                STREAMS.wait_value(eid, v)
            result = call.signature.func(*args, **kwargs)
    except BaseException as e:  # noqa: BLE001 - transported as a value
        elapsed = time.perf_counter() - t0
        METRICS.observe("lzy_op_run", elapsed)
        METRICS.inc("lzy_op_failures", op=name)
        tb = traceback.format_exc()
        # exception as a first-class output (reference startup.py:124-169)
        snap.put(call.exception_id, (type(e).__name__, str(e), tb))
        raise LzyExecutionError(
            f"Op {name} failed: {type(e).__name__}: {e}", task_id=call.id,
            remote_traceback=tb,
        ) from e
    finally:
        capture.unroute_current_thread()
        _archive_logs(call, out_buf.getvalue(), err_buf.getvalue())

    elapsed = time.perf_counter() - t0
    METRICS.observe("lzy_op_run", elapsed)
    METRICS.inc("lzy_op_runs", op=name)

    _store_outputs(call, result, stream=stream)

    if call.cache:
        for eid in call.entry_ids:
            snap.persist(eid)

    FAULTS.hit("executor.after_run")


def _archive_logs(call: "LzyCall", out: str, err: str) -> None:
    """Persist captured op std-logs beside the run's storage prefix.

    Reference capability: worker stdout/stderr flow to a Kafka topic and
    are archived to S3 by s3-sink (reference: lzy/s3-sink Job.java:38-86);
    here the archive is one write to the durable tier per op that logged.
    """
    if not out and not err:
        return
    try:
        wf = call.workflow
        base = (
            f"{wf.owner.storage_uri}/lzy_logs/{wf.execution_id}"
            f"/{call.callable_name}-{call.id[:8]}"
        )
        if out:
            wf.snapshot.storage.write_bytes(base + ".out", out.encode())
        if err:
            wf.snapshot.storage.write_bytes(base + ".err", err.encode())
    except Exception:  # noqa: BLE001 - archiving must never fail the op
        _LOG.warning("failed to archive logs for %s", call.callable_name)


def _input_value(call: "LzyCall", entry_id: str, typ: type) -> Any:
    snap = call.workflow.snapshot
    if call.lazy_arguments:
        return lzy_proxy(entry_id, (typ,), call.workflow)
    got = snap.try_get(entry_id)
    if not got.found:
        raise LzyExecutionError(
            f"Input entry {entry_id} for op {call.callable_name} has no value",
            task_id=call.id,
        )
    value = got.value
    return materialize(value) if is_lzy_proxy(value) else value


def _store_outputs(call: "LzyCall", result: Any, stream=None) -> None:
    from lzy_amd.runtime.streams import STREAMS

    snap = call.workflow.snapshot
    n = len(call.entry_ids)
    if n == 1:
        outputs: Tuple[Any, ...] = (result,)
    else:
        if not isinstance(result, tuple) or len(result) != n:
            raise LzyExecutionError(
                f"Op {call.callable_name} declared {n} outputs but returned "
                f"{type(result).__name__} of length "
                f"{len(result) if isinstance(result, (tuple, list)) else 'n/a'}",
                task_id=call.id,
            )
        outputs = result
    for eid, value in zip(call.entry_ids, outputs):
        value = materialize(value) if is_lzy_proxy(value) else value
        STREAMS.record_output(eid, value, stream=stream)  # pre-publication
        snap.put(eid, value)
