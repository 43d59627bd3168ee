"""GpuPoolRuntime: one process per MI355X GPU, driven by an in-process scheduler.

This is the MI355X-native collapse of the reference's entire service fleet
(reference layers L3-L5: lzy-service ExecuteGraph pipeline, graph-executor-2
ExecuteTaskAction, scheduler job chains, allocator VM lifecycle — SURVEY.md
§3.2): on one 8-GPU node there are no VMs to allocate and no conda to sync,
so "allocation" is picking a rank, "scheduling" is a ready-frontier walk of
the C++ DAG core, and dispatch is a ~50 us control message instead of a
1 s tick + K8s pod create.

Topology:
  * rank 0 = driver: runs the user workflow script, owns the scheduler,
    and is ALSO a worker (its GPU is not idle).
  * ranks 1..N-1 = workers: enter the serve loop at pool init and execute
    tasks / transfers until shutdown (then sys.exit(0)).
  * control plane: one AF_UNIX connection per worker to the driver (star).
  * data plane: torch.distributed isend/irecv on a dedicated process
    group — RCCL over xGMI for device tensors, gloo for CPU bytes;
    transfer commands carry a driver-assigned global sequence so every
    rank issues its p2p ops in a pairwise-consistent order (RCCL matches
    by order, not tags — no deadlocks by construction).
  * gang ops (@op(gpu_count=k)): the same TaskSpec dispatched to k ranks
    with a shared RCCL subgroup for in-op collectives/DDP; gangs with
    disjoint rank sets run concurrently.
  * streamed merge-tree plans (@op(pair_reduce=...)): connected merge
    components execute as one chunk-pipelined p2p collective on a
    dedicated process group (channels/treeplan.py).
  * failure model: mid-flight StopGraph (abort/Ctrl-C cancels queued
    tasks, bounded drain); worker death re-dispatches inflight tasks to
    survivors with inputs re-rooted from live owners or the durable
    tier; transfer-source death fails consumers fast (bounded settle
    waits, per-transfer gloo tags) and retries through the same
    machinery.  Dead ranks stay excluded from all later placement.
"""
from __future__ import annotations

import atexit
import logging
import os
import queue
import sys
import tempfile
import threading
import time
from typing import TYPE_CHECKING, Any, Dict, List, Optional, Sequence, Set, Tuple

import torch
import torch.distributed as dist

from lzy_amd.channels.control import DriverControl, WorkerControl, broadcast_address
from lzy_amd.channels.transport import (
    KIND_BYTES,
    KIND_TENSOR,
    EntryMeta,
    Transport,
    describe_value,
    pickle_value,
    unpickle_value,
)
from lzy_amd.exceptions import BadProvisioningError, LzyExecutionError
from lzy_amd.runtime.base import Runtime
from lzy_amd.runtime.taskspec import TaskResult, TaskSpec, WorkerStore, run_taskspec
from lzy_amd.sched import Dag, Journal
from lzy_amd.serialization.registry import LzySerializerRegistry
from lzy_amd.storage.api import StorageConfig
from lzy_amd.storage.fs import FsStorageClient
from lzy_amd.utils.logs import OpLogCapture
from lzy_amd.utils.metrics import METRICS

if TYPE_CHECKING:
    from lzy_amd.core.call import LzyCall
    from lzy_amd.core.workflow import LzyWorkflow

_LOG = logging.getLogger("lzy_amd.pool")

INLINE_LIMIT = 256 << 10  # CPU values up to 256 KiB travel inside the TaskSpec

# function -> cloudpickle bytes memo (driver side; ops are module-level
# callables so identity is stable and re-pickling per dispatch is waste)
_FUNC_BYTES_CACHE: Dict[int, Tuple[Any, bytes]] = {}


def _func_bytes(func) -> bytes:
    key = id(func)
    hit = _FUNC_BYTES_CACHE.get(key)
    if hit is not None and hit[0] is func:
        return hit[1]
    data = pickle_value(func)
    _FUNC_BYTES_CACHE[key] = (func, data)
    return data


def _storage_root() -> str:
    from lzy_amd.config import get_config

    root = get_config().storage or os.path.join(
        tempfile.gettempdir(), "lzy_amd_storage"
    )
    return root[len("file://"):] if root.startswith("file://") else root


class WorkerAgent:
    """Per-rank serve loop + executor thread (every rank, driver included)."""

    def __init__(self, rank: int, world: int, address: str, pg_data, device,
                 pg_stream=None):
        self.rank = rank
        self.world = world
        self.device = device
        self.pg_stream = pg_stream
        self.store = WorkerStore(device=device)
        self.serializers = LzySerializerRegistry()
        self.storage = FsStorageClient()
        self.transport = Transport(pg_data, device, world=world)
        self.ctrl = WorkerControl(rank, address)
        self._pending: Dict[str, Tuple[list, Any]] = {}  # entry -> (works, fin)
        self._outbox: List[Tuple[list, Any]] = []
        self._groups: Dict[str, Any] = {}
        self._exec_q: "queue.Queue[dict]" = queue.Queue()
        self._shutdown = False
        # StopGraph support: task ids cancelled mid-flight (reference:
        # AbortExecution stopGraphs, workflow-service.proto StopGraph).
        # Queued tasks check this when an executor thread picks them up.
        self._cancelled: Dict[str, str] = {}
        # small task pool per rank: independent tasks of one rank overlap
        # (each executor thread gets its own HIP stream via
        # runtime/streams.py).  Gang tasks are admitted DRIVER-side only
        # onto rank sets disjoint from every inflight gang, so collective
        # order can never diverge on any rank.
        from concurrent.futures import ThreadPoolExecutor

        from lzy_amd.config import get_config

        n_exec = max(1, int(getattr(get_config(), "exec_threads", 2)))
        self._task_pool = ThreadPoolExecutor(
            max_workers=n_exec, thread_name_prefix=f"lzy-task-r{rank}"
        )
        # streamed plans are communication-bound and sit on the makespan
        # critical path: a dedicated thread keeps them from queueing
        # behind compute tasks (plans are serialized pool-wide anyway)
        self._plan_pool = ThreadPoolExecutor(
            max_workers=1, thread_name_prefix=f"lzy-plan-r{rank}"
        )
        self._futs_lock = threading.Lock()
        self._futs: Set[Any] = set()
        self._exec_thread = threading.Thread(
            target=self._executor, daemon=True, name=f"lzy-exec-r{rank}"
        )
        self._exec_thread.start()
        # live log streaming from ranks > 0 (reference: worker→Kafka→
        # client ReadStdSlots stream while the op runs): a flusher ships
        # newline-bounded buffer deltas to the driver periodically
        self._livelogs: Dict[str, list] = {}  # tid -> [name, out, err, so, se]
        self._livelog_lock = threading.Lock()
        if rank != 0:
            self._log_period = float(
                getattr(get_config(), "log_stream_period_s", 0.25)
            )
            threading.Thread(
                target=self._livelog_flusher, daemon=True,
                name=f"lzy-logflush-r{rank}",
            ).start()
            # periodic liveness beacon (reference: AllocatorAgent
            # heartbeat TimerTask, allocator-api AllocatorAgent.java:26).
            # Socket EOF already detects process DEATH; heartbeats let
            # the driver flag a HUNG-but-alive worker.
            self._hb_period = float(
                getattr(get_config(), "heartbeat_period_s", 2.0)
            )
            threading.Thread(
                target=self._heartbeat, daemon=True,
                name=f"lzy-hb-r{rank}",
            ).start()
        OpLogCapture.instance().install()

    # -- serve loop ---------------------------------------------------------

    def serve_forever(self) -> None:
        while not self._shutdown:
            try:
                msg = self.ctrl.recv()
            except (EOFError, OSError):
                break
            try:
                self._handle(msg)
            except BaseException as e:  # noqa: BLE001 - serve loop survives
                _LOG.exception("agent r%d serve error on %s", self.rank,
                               msg.get("cmd"))
                self._report_error(msg, e)

    def handle_local(self, msg: dict) -> None:
        """Inline command path for the driver-process rank 0 (no socket
        hop); same error reporting as the serve loop."""
        try:
            self._handle(msg)
        except BaseException as e:  # noqa: BLE001 - must not kill the caller
            _LOG.exception("agent r%d local-handle error on %s", self.rank,
                           msg.get("cmd"))
            self._report_error(msg, e)

    def _handle(self, msg: dict) -> None:
        cmd = msg["cmd"]
        if cmd == "task":
            self._exec_q.put(msg)
        elif cmd == "xfer_send_batch":
            ops_all: list = []
            keeps: list = []
            failed_items: list = []
            for item in msg["items"]:
                eid = item["entry"]
                try:
                    if not self.store.has(eid):  # settle may be landing
                        # config-derived (settle_wait_s, cached by the
                        # store): a multi-GB fan-in over one xGMI link
                        # can legitimately outlast a fixed small bound
                        self.store.wait_present(eid)
                    value = self.store.get(eid)
                except (KeyError, RuntimeError) as e:
                    # this rank never (or no longer) holds the entry —
                    # e.g. its own inbound copy died with the source.
                    # Tell the driver so it can poison the waiting
                    # consumer instead of letting it block on a recv
                    # that will never be fed.
                    failed_items.append((item, f"{type(e).__name__}: {e}"))
                    continue
                ops, keep = self.transport.send_ops(
                    value, self.store.pickled.get(eid), item["dst"],
                    tag=item.get("tag", 0),
                )
                ops_all += ops
                keeps.append(keep)
            works = Transport.issue(ops_all)
            self._outbox.append((works, keeps))
            self._prune_outbox()
            for item, err in failed_items:
                self.ctrl.send_event({
                    "ev": "xfer_failed", "rank": self.rank,
                    "entry": item["entry"], "dst": item["dst"],
                    "error": err,
                })
        elif cmd == "xfer_recv_batch":
            ops_all = []
            fins: list = []
            for item in msg["items"]:
                meta = EntryMeta.from_wire(item["meta"])
                ops, fin = self.transport.recv_ops(
                    meta, item["src"], tag=item.get("tag", 0)
                )
                fins.append((item["entry"], len(ops), fin))
                ops_all += ops
            works = Transport.issue(ops_all)
            i = 0
            for eid, nops, fin in fins:
                self._pending[eid] = (works[i: i + nops], fin)
                i += nops
        elif cmd == "ipc_export":
            from lzy_amd.channels.transport import export_ipc

            data = export_ipc(self.store.get(msg["entry"]))
            self.ctrl.send_event(
                {"ev": "ack", "tag": msg["tag"], "rank": self.rank,
                 "payload": data}
            )
        elif cmd == "ipc_export_batch":
            from lzy_amd.channels.transport import export_ipc

            handles = {
                eid: export_ipc(self.store.get(eid)) for eid in msg["entries"]
            }
            self.ctrl.send_event(
                {"ev": "ack", "tag": msg["tag"], "rank": self.rank,
                 "payload": handles}
            )
        elif cmd == "ipc_import":
            from lzy_amd.channels.transport import import_ipc

            data = msg["data"]
            dev = self.device
            self._pending[msg["entry"]] = (
                None, (lambda d=data: import_ipc(d, dev))
            )
        elif cmd == "settle":
            self._exec_q.put(msg)
        elif cmd == "stream_plan":
            self._exec_q.put(msg)
        elif cmd == "new_group":
            ranks = msg["ranks"]
            tag = msg["tag"]
            if tag not in self._groups:
                self._groups[tag] = dist.new_group(ranks=ranks)
            self.ctrl.send_event({"ev": "ack", "tag": tag, "rank": self.rank})
        elif cmd == "barrier":
            self._exec_q.put(msg)
        elif cmd == "preflight":
            self._exec_q.put(msg)
        elif cmd == "load_serializers":
            self.serializers.load_user_serializers(msg["payload"])
            self.ctrl.send_event({"ev": "ack", "tag": msg["tag"], "rank": self.rank})
        elif cmd == "cancel_tasks":
            reason = msg.get("reason", "graph stopped")
            for tid in msg["ids"]:
                self._cancelled[tid] = reason
        elif cmd == "poison":
            for eid in msg["entries"]:
                self.store.poison(eid, msg.get("reason", "producer failed"))
        elif cmd == "drop_entries":
            for eid in msg["entries"]:
                self.store.drop(eid)
        elif cmd == "clear_store":
            self.store.clear()
            self._cancelled.clear()
        elif cmd == "shutdown":
            self._shutdown = True
            self._exec_q.put({"cmd": "_stop"})

    def run_preflight(self) -> None:
        """Exercise every communicator once (default, pg_data, pg_stream;
        CPU and, on the RCCL path, CUDA) so a broken rendezvous fails
        fast with a rank-tagged error instead of hanging mid-benchmark.
        All ranks must run this concurrently (collectives)."""
        groups = [("default", None), ("pg_data", self.transport._pg),
                  ("pg_stream", self.pg_stream)]
        for name, group in groups:
            t = torch.ones(1)
            dist.all_reduce(t, group=group)
            if int(t.item()) != self.world:
                raise RuntimeError(
                    f"preflight {name}: CPU all_reduce gave {t.item()}, "
                    f"want {self.world}"
                )
            if self.device is not None and self.transport._cuda_p2p:
                d = torch.ones(1, device=self.device)
                dist.all_reduce(d, group=group)
                torch.cuda.synchronize(self.device)
                if int(d.item()) != self.world:
                    raise RuntimeError(
                        f"preflight {name}: CUDA all_reduce gave {d.item()}, "
                        f"want {self.world}"
                    )

    def _prune_outbox(self) -> None:
        still = []
        for works, keep in self._outbox:
            if not all(w.is_completed() for w in works):
                still.append((works, keep))
        self._outbox = still

    # -- executor thread ----------------------------------------------------

    def _executor(self) -> None:
        while True:
            msg = self._exec_q.get()
            cmd = msg["cmd"]
            if cmd == "_stop":
                self._task_pool.shutdown(wait=False)
                self._plan_pool.shutdown(wait=False)
                return
            if cmd in ("task", "settle", "stream_plan"):
                pool = (
                    self._plan_pool if cmd == "stream_plan"
                    else self._task_pool
                )
                fut = pool.submit(self._run_guarded, msg)
                with self._futs_lock:
                    self._futs.add(fut)
                fut.add_done_callback(self._fut_done)
            elif cmd == "preflight":
                try:
                    self.run_preflight()
                    self.ctrl.send_event(
                        {"ev": "ack", "tag": msg["tag"], "rank": self.rank}
                    )
                except BaseException as e:  # noqa: BLE001
                    self._report_error(msg, e)
            elif cmd == "barrier":
                # a barrier orders after every previously submitted task
                self._drain_tasks()
                try:
                    if self.device is not None:
                        torch.cuda.synchronize(self.device)
                    if dist.is_initialized():
                        # CPU all-reduce = gloo path: correct under GPU
                        # oversubscription and free of RCCL rendezvous
                        dist.all_reduce(torch.zeros(1))
                    if self.device is not None:
                        torch.cuda.synchronize(self.device)
                    self.ctrl.send_event(
                        {
                            "ev": "barrier_done",
                            "tag": msg["tag"],
                            "rank": self.rank,
                            "ts": time.perf_counter(),
                        }
                    )
                except BaseException as e:  # noqa: BLE001
                    self._report_error(msg, e)

    def _fut_done(self, fut) -> None:
        with self._futs_lock:
            self._futs.discard(fut)

    def _drain_tasks(self) -> None:
        while True:
            with self._futs_lock:
                futs = list(self._futs)
            if not futs:
                return
            for f in futs:
                f.exception()  # wait; errors were already reported

    def _run_guarded(self, msg: dict) -> None:
        cmd = msg["cmd"]
        spec = msg.get("spec")
        if spec is not None and spec.task_id in self._cancelled:
            # cancelled while queued: skip the op entirely (the running
            # ones can't be killed — this is what bounds abort latency
            # to ONE op duration instead of the whole queue)
            self.ctrl.send_event({
                "ev": "task_cancelled", "task_id": spec.task_id,
                "rank": self.rank,
            })
            return
        try:
            if cmd == "task":
                self._run_task(msg)
            elif cmd == "stream_plan":
                self._run_stream_plan(msg)
            else:
                self._settle(msg["entries"])
                self.ctrl.send_event(
                    {"ev": "settled", "tag": msg["tag"], "rank": self.rank}
                )
        except BaseException as e:  # noqa: BLE001 - agent must survive
            _LOG.exception("agent r%d failed handling %s", self.rank, cmd)
            self._report_error(msg, e)

    def _run_stream_plan(self, msg: dict) -> None:
        """Execute this rank's schedule of a streamed merge-tree plan
        (channels/treeplan.py); p2p runs on the dedicated pg_stream."""
        from lzy_amd.channels.treeplan import run_stream_plan

        plan = msg["plan"]
        if plan["plan_id"] in self._cancelled:
            self.ctrl.send_event({
                "ev": "plan_failed", "plan_id": plan["plan_id"],
                "rank": self.rank, "error": "cancelled",
            })
            return
        try:
            results = run_stream_plan(
                plan, msg["steps"], self.store,
                self.pg_stream, self.device,
                wait_timeout=msg.get("wait_timeout", 120.0),
            )
        except BaseException as e:  # noqa: BLE001
            _LOG.exception("agent r%d stream plan failed", self.rank)
            self.ctrl.send_event({
                "ev": "plan_failed", "plan_id": plan["plan_id"],
                "rank": self.rank, "error": f"{type(e).__name__}: {e}",
            })
            return
        self.ctrl.send_event({
            "ev": "plan_done", "plan_id": plan["plan_id"],
            "rank": self.rank, "outputs": results,
        })

    def _report_error(self, msg: dict, e: BaseException) -> None:
        spec = msg.get("spec")
        where = f" in op {spec.name}" if spec is not None else f" handling {msg.get('cmd')}"
        self.ctrl.send_event(
            {
                "ev": "agent_error",
                "rank": self.rank,
                "error": f"{type(e).__name__}: {e}{where}",
                "task_id": spec.task_id if spec is not None else None,
            }
        )

    def _settle(self, entries: Sequence[str]) -> None:
        """Complete pending transfers, landing values in the store."""
        for eid in entries:
            pending = self._pending.pop(eid, None)
            if pending is None:
                # either already settled, or a concurrent task's settle
                # owns the transfer: wait for the value to land
                if not self.store.has(eid) and not self.store.wait_present(eid):
                    raise RuntimeError(
                        f"rank {self.rank}: entry {eid} never arrived"
                    )
                continue
            works, fin = pending
            if works:
                # gloo waits are bounded: a transfer whose sender died or
                # never sent must fail this task promptly, not hang the
                # executor forever.  RCCL waits are stream-enqueue
                # (non-host-blocking) — no timeout semantics needed.
                import datetime as _dt

                bounded = not self.transport._cuda_p2p
                tmo = _dt.timedelta(seconds=self.store._settle_wait)
                for w in works:
                    if bounded:
                        w.wait(tmo)
                    else:
                        w.wait()
            value = fin()
            pickled = None
            if not isinstance(value, torch.Tensor):
                pickled = self.store.pickled.get(eid)
            # RCCL recv completion is stream-ordered, not host-ordered:
            # publish an event (pre-publication) so consumer op streams
            # order after it
            from lzy_amd.runtime.streams import STREAMS

            STREAMS.record_output(eid, value)
            self.store.put(eid, value, pickled=pickled)

    # -- live log streaming --------------------------------------------------

    def _live_sink_for(self, spec: "TaskSpec"):
        """Registration hook handed to run_taskspec on ranks > 0."""
        if self.rank == 0:
            return None  # echoed live on the shared console already

        def register(out_buf, err_buf):
            tid = spec.task_id
            with self._livelog_lock:
                self._livelogs[tid] = [spec.name, out_buf, err_buf, 0, 0]

            def unregister():
                self._flush_livelog(tid, final=True)
                with self._livelog_lock:
                    self._livelogs.pop(tid, None)

            return unregister

        return register

    def _flush_livelog(self, tid: str, final: bool = False) -> None:
        # the whole read-compute-update-send runs under ONE lock: the
        # periodic flusher and the op thread's final flush would
        # otherwise race to ship the same delta twice (and could
        # reorder chunks on the wire)
        with self._livelog_lock:
            rec = self._livelogs.get(tid)
            if rec is None:
                return
            name, out_buf, err_buf, so, se = rec

            def delta(buf, shipped):
                s = buf.getvalue()[shipped:]
                if not final:
                    # hold partial lines: the driver prints whole lines
                    cut = s.rfind("\n") + 1
                    s = s[:cut]
                return s

            d_out = delta(out_buf, so)
            d_err = delta(err_buf, se)
            if not d_out and not d_err:
                return
            rec[3] += len(d_out)
            rec[4] += len(d_err)
            try:
                self.ctrl.send_event({
                    "ev": "log_chunk", "task_id": tid, "name": name,
                    "rank": self.rank, "out": d_out, "err": d_err,
                })
            except (OSError, BrokenPipeError):
                pass

    def _livelog_flusher(self) -> None:
        while not self._shutdown:
            time.sleep(self._log_period)
            with self._livelog_lock:
                tids = list(self._livelogs)
            for tid in tids:
                self._flush_livelog(tid)

    def _heartbeat(self) -> None:
        while not self._shutdown:
            time.sleep(self._hb_period)
            try:
                self.ctrl.send_event({"ev": "heartbeat", "rank": self.rank})
            except (OSError, BrokenPipeError):
                return

    def _run_task(self, msg: dict) -> None:
        spec: TaskSpec = msg["spec"]
        self._settle(spec.wait_entries)
        if self.device is not None:
            torch.cuda.set_device(self.device)
        gang_group = None
        if spec.gang is not None:
            tag = spec.gang["tag"]
            gang_group = self._groups.get(tag)  # None -> default pg
        result = run_taskspec(
            spec, self.store, self.serializers, self.storage,
            gang_group=gang_group,
            # rank 0 echoes op logs on its own console; other ranks
            # stream them live (log_chunk events) and ship the full
            # capture in the TaskResult (the console belongs to the client)
            echo_logs=(self.rank == 0),
            live_sink=self._live_sink_for(spec),
        )
        self.ctrl.send_event(
            {
                "ev": "task_done" if result.ok else "task_failed",
                "rank": self.rank,
                "result": result,
            }
        )


class GpuPool:
    """Process-wide pool singleton."""

    _instance: Optional["GpuPool"] = None

    def __init__(self) -> None:
        self.rank = int(os.environ.get("RANK", "0"))
        self.world = int(os.environ.get("WORLD_SIZE", "1"))
        self.is_driver = self.rank == 0
        self.device: Optional[torch.device] = None
        self.driver_ctrl: Optional[DriverControl] = None
        self.agent: Optional[WorkerAgent] = None
        self.events: "queue.Queue[Tuple[int, dict]]" = queue.Queue()
        self._acks: Dict[str, Set[int]] = {}
        self._ack_payloads: Dict[str, Dict[int, Any]] = {}
        self._ack_cv = threading.Condition()
        self._group_tags: Set[str] = set()
        self._seq = 0
        # liveness: last event (heartbeat or otherwise) per rank
        self.hb_last: Dict[int, float] = {}
        self._stall_flagged: Set[int] = set()
        self.dead_ranks: Set[int] = set()

    @classmethod
    def get(cls) -> "GpuPool":
        if cls._instance is None:
            cls._instance = GpuPool()
            cls._instance._init()
        return cls._instance

    def _init(self) -> None:
        if torch.cuda.is_available():
            local = int(os.environ.get("LOCAL_RANK", self.rank))
            self.device = torch.device("cuda", local % torch.cuda.device_count())
            torch.cuda.set_device(self.device)

        # one rank per physical GPU is the production shape; ranks may
        # exceed GPUs only in test harnesses, where RCCL cannot build a
        # comm and the data plane stages through gloo instead
        cuda_ok = (
            self.device is not None
            and torch.cuda.device_count() >= self.world
        )
        pg_data = None
        if self.world > 1:
            if not dist.is_initialized():
                # explicit composite: CPU collectives on gloo (control
                # broadcasts, barriers), CUDA tensors on nccl (=RCCL)
                import datetime as _dt

                backend = "cpu:gloo,cuda:nccl" if torch.cuda.is_available() else "gloo"
                dist.init_process_group(
                    backend=backend,
                    rank=self.rank,
                    world_size=self.world,
                    device_id=self.device if cuda_ok else None,
                    # fail fast instead of the 30-min default: a broken
                    # rendezvous should error, not hang the harness
                    timeout=_dt.timedelta(seconds=180),
                )
            pg_data = dist.new_group()  # dedicated transfer group
            # streamed tree plans issue p2p from EXECUTOR threads; their
            # own group keeps that traffic's issue order independent of
            # the serve-loop-issued transfers on pg_data (separate comm =
            # separate match space).  Created here so every rank joins
            # the (eager, ncclCommSplit under device_id) creation.
            self.pg_stream = dist.new_group()

        if self.is_driver:
            self.driver_ctrl = DriverControl(self.world, self._on_event)
            address = self.driver_ctrl.address
            accept_thread = threading.Thread(
                target=self.driver_ctrl.accept_all, daemon=True, name="lzy-accept"
            )
            accept_thread.start()
        else:
            address = None
        if self.world > 1:
            address = broadcast_address(address)

        self.agent = WorkerAgent(
            self.rank, self.world, address, pg_data, self.device,
            pg_stream=getattr(self, "pg_stream", None),
        )

        if self.is_driver:
            accept_thread.join()
            if self.world > 1:
                threading.Thread(
                    target=self._stall_monitor, daemon=True,
                    name="lzy-stallmon",
                ).start()
            # short-circuit the loopback: rank 0 lives in this process,
            # so its commands and events skip the AF_UNIX socket (two
            # pickle hops saved per task/transfer/event).  The serve
            # thread stays up but idles — the socket goes silent.
            self.driver_ctrl.local_handler = self.agent.handle_local
            self.agent.ctrl.local_sink = self._on_event
            serve_thread = threading.Thread(
                target=self.agent.serve_forever, daemon=True, name="lzy-agent-serve"
            )
            serve_thread.start()
            atexit.register(self.shutdown)
        else:
            # workers serve until shutdown, then exit the process; the
            # user script body after pool entry never runs on workers.
            self.agent.serve_forever()
            if dist.is_initialized():
                try:
                    dist.destroy_process_group()
                except Exception:
                    pass
            sys.exit(0)

    # -- driver-side event plumbing -----------------------------------------

    def _on_event(self, rank: int, msg: dict) -> None:
        ev = msg.get("ev")
        self.hb_last[rank] = time.monotonic()  # any event proves liveness
        if ev == "heartbeat":
            return
        if ev == "worker_lost":
            self.dead_ranks.add(rank)
        if ev in ("ack", "settled"):
            with self._ack_cv:
                key = f"{ev}:{msg['tag']}"
                self._acks.setdefault(key, set()).add(rank)
                if "payload" in msg:
                    self._ack_payloads.setdefault(key, {})[rank] = msg["payload"]
                self._ack_cv.notify_all()
        else:
            self.events.put((rank, msg))

    def wait_acks(self, kind: str, tag: str, ranks: Sequence[int],
                  timeout: Optional[float] = None) -> Dict[int, Any]:
        if timeout is None:
            from lzy_amd.config import get_config

            timeout = float(getattr(get_config(), "ack_wait_s", 300.0))
        key = f"{kind}:{tag}"
        want = set(ranks)
        with self._ack_cv:
            ok = self._ack_cv.wait_for(
                lambda: want.issubset(self._acks.get(key, set())), timeout
            )
            if not ok:
                raise TimeoutError(f"waiting for {kind} {tag} from {ranks}")
            self._acks.pop(key, None)
            return self._ack_payloads.pop(key, {})

    def live_ranks(self):
        return [r for r in range(self.world) if r not in self.dead_ranks]

    def next_seq(self) -> int:
        self._seq += 1
        return self._seq

    def ensure_group(self, ranks: Sequence[int]) -> str:
        """All ranks must call dist.new_group with the same list — broadcast
        the creation command and wait for every rank's ack."""
        ranks = sorted(ranks)
        if len(ranks) == self.world:
            return "__default__"
        tag = ",".join(map(str, ranks))
        if tag not in self._group_tags:
            if self.dead_ranks:
                # dist.new_group is collective over the DEFAULT group —
                # a dead rank can never join, so creation would hang.
                # Cached groups keep working; new ones cannot exist.
                raise RuntimeError(
                    f"cannot create process group {ranks}: rank(s) "
                    f"{sorted(self.dead_ranks)} are dead (new_group is "
                    "collective over all ranks)"
                )
            self.driver_ctrl.broadcast({"cmd": "new_group", "ranks": ranks, "tag": tag})
            self.wait_acks("ack", tag, range(self.world))
            self._group_tags.add(tag)
        return tag

    def _stall_monitor(self) -> None:
        """Flag workers that stopped heartbeating but whose socket is
        still open (hung in a collective, deadlocked op, stuck driver):
        a warning + metric, never a kill — the reference's heartbeat-miss
        policy (VM declared dead) is socket-EOF's job here; a hung-but-
        alive rank still owns comms and killing it would poison them."""
        from lzy_amd.config import get_config

        period = float(getattr(get_config(), "heartbeat_period_s", 2.0))
        while self.driver_ctrl is not None:
            time.sleep(period)
            now = time.monotonic()
            for r in range(1, self.world):
                if r in self.dead_ranks:
                    continue  # dead, not hung — already handled
                last = self.hb_last.get(r)
                if last is None:
                    continue
                silent = now - last
                if silent > 4 * period and r not in self._stall_flagged:
                    self._stall_flagged.add(r)
                    METRICS.inc("lzy_worker_stalls")
                    _LOG.warning(
                        "worker rank %d silent for %.1fs (socket alive) — "
                        "possibly hung", r, silent,
                    )
                elif silent <= 4 * period:
                    self._stall_flagged.discard(r)

    def preflight(self) -> None:
        """Driver-side: run the communicator preflight on every rank
        (fail-fast RCCL/gloo diagnostics — all ranks join the
        collectives; a failure surfaces as a rank-tagged TimeoutError or
        agent error instead of a silent mid-benchmark hang)."""
        if self.world <= 1 or not self.is_driver:
            return
        tag = f"pf{self.next_seq()}"
        self.driver_ctrl.broadcast({"cmd": "preflight", "tag": tag})
        self.wait_acks("ack", tag, self.live_ranks(), timeout=240.0)

    def sync_all(self) -> Dict[int, float]:
        """Barrier across all ranks (through exec queues, so it orders after
        all dispatched tasks); returns per-rank completion timestamps."""
        tag = f"b{self.next_seq()}"
        self.driver_ctrl.broadcast({"cmd": "barrier", "tag": tag})
        deadline = time.monotonic() + 600
        want = self.world - len(self.dead_ranks)
        ts: Dict[int, float] = {}
        stash = []
        while len(ts) < want:
            if time.monotonic() > deadline:
                raise TimeoutError("pool barrier timed out")
            try:
                rank, msg = self.events.get(timeout=1.0)
            except queue.Empty:
                continue
            if msg.get("ev") == "barrier_done" and msg.get("tag") == tag:
                ts[rank] = msg["ts"]
            else:
                stash.append((rank, msg))  # foreign event; re-deliver after
        for item in stash:
            self.events.put(item)
        return ts

    def shutdown(self) -> None:
        if self.driver_ctrl is not None:
            try:
                self.driver_ctrl.broadcast({"cmd": "shutdown"})
                time.sleep(0.2)
                self.driver_ctrl.close()
            except Exception:
                pass
            self.driver_ctrl = None
            if dist.is_initialized():
                try:
                    dist.destroy_process_group()
                except Exception:
                    pass


class GpuPoolRuntime(Runtime):
    """Driver-side Runtime implementation over the pool."""

    def __init__(self, journal_dir: Optional[str] = None):
        self._journal_dir = journal_dir or os.path.join(
            tempfile.gettempdir(), "lzy_amd_journal"
        )
        # opportunistic GC at runtime construction (reference: the
        # lzy-service GarbageCollector sweeps stale executions
        # periodically; one sweep per runtime keeps the journal dir
        # bounded without a daemon)
        try:
            from lzy_amd.config import get_config
            from lzy_amd.storage.gc import gc_journals

            ttl_h = float(getattr(get_config(), "journal_ttl_hours", 168.0))
            if ttl_h > 0:
                gc_journals(self._journal_dir, ttl_seconds=ttl_h * 3600.0)
        except Exception:  # noqa: BLE001 - GC must never block startup
            pass
        self._journal: Optional[Journal] = None
        self._pool: Optional[GpuPool] = None
        # single-flight: the driver scheduler state is per-workflow.
        # threading.Lock has no ownership, so finish/abort track WHICH
        # workflow holds the flight and only release for that one — a
        # stray abort on a never-started workflow must not unlock a
        # different thread's active flight.
        self._flight = threading.Lock()
        self._flight_owner: Optional[str] = None
        self._flight_guard = threading.Lock()  # atomic owner check+clear
        self._active_sched: Optional["_DriverScheduler"] = None

    @property
    def pool(self) -> GpuPool:
        if self._pool is None:
            self._pool = GpuPool.get()
        return self._pool

    def storage(self) -> Optional[StorageConfig]:
        return StorageConfig(uri=f"file://{_storage_root()}")

    def start(self, workflow: "LzyWorkflow") -> None:
        from lzy_amd.utils.metrics import timed

        self._flight.acquire()  # one workflow at a time per runtime
        self._flight_owner = workflow.execution_id
        with timed("lzy_wf_start"):
            self._start(workflow)

    def _start(self, workflow: "LzyWorkflow") -> None:
        pool = self.pool  # workers never get past this line (serve loop)
        assert pool.is_driver
        # the rank-0 agent store backs the workflow snapshot: captured args
        # are instantly "on" rank 0, and fetched outputs appear in the
        # snapshot without copies.
        workflow.snapshot._values = pool.agent.store.values
        workflow._entry_meta = {}

        def _fetcher(entry_id: str) -> None:
            meta = workflow._entry_meta.get(entry_id)
            if meta is not None and meta.owners and 0 not in meta.owners:
                self.fetch_entry(entry_id, meta)

        workflow.snapshot.fetcher = _fetcher
        self._journal = Journal(
            os.path.join(self._journal_dir, f"{workflow.execution_id}.jsonl")
        )
        payload = workflow.owner.serializer_registry.user_serializers_payload()
        # ship user serializers only when they changed (usually: never)
        if payload != getattr(pool, "_last_ser_payload", None):
            tag = f"ser{pool.next_seq()}"
            pool.driver_ctrl.broadcast(
                {"cmd": "load_serializers", "payload": payload, "tag": tag}
            )
            pool.wait_acks("ack", tag, pool.live_ranks())
            pool._last_ser_payload = payload

    def exec(self, workflow: "LzyWorkflow", calls: Sequence["LzyCall"]) -> None:
        from lzy_amd.utils.metrics import timed

        # leaf tracking for finish-time persistence: results a later op
        # never consumed may still be read AFTER the workflow exits
        # (reference: everything lands on S3; tutorial 3 prints an op
        # result outside the block)
        produced = workflow.__dict__.setdefault("_produced_entries", set())
        consumed = workflow.__dict__.setdefault("_consumed_entries", set())
        for c in calls:
            produced.update(c.entry_ids)
            consumed.update(c.input_entry_ids())

        with timed("lzy_wf_exec"):
            sched = _DriverScheduler(self.pool, workflow, calls, self._journal)
            self._active_sched = sched
            try:
                sched.run()
            finally:
                self._active_sched = None

    def finish(self, workflow: "LzyWorkflow") -> None:
        from lzy_amd.utils.metrics import timed

        try:
            self._finish_inner(workflow)
        finally:
            # atomic check-and-clear: a client-thread finish racing a
            # side-thread abort must release exactly once — a double
            # release could free the NEXT workflow's freshly acquired
            # flight
            with self._flight_guard:
                owned = self._flight_owner == workflow.execution_id
                if owned:
                    self._flight_owner = None
            if owned:
                try:
                    self._flight.release()
                except RuntimeError:
                    pass

    def _finish_inner(self, workflow: "LzyWorkflow") -> None:
        from lzy_amd.utils.metrics import timed

        if workflow._snapshot is None:
            return  # never started: nothing to drop/close
        with timed("lzy_wf_finish"):
            # never-consumed results may still be read after the block
            # exits (reference tutorial 3 prints an op result outside
            # the workflow; on S3 they simply persist) — keep the LEAF
            # values hot and let post-exit materialization fetch them;
            # everything consumed is dropped for HBM pressure control
            produced = workflow.__dict__.get("_produced_entries", set())
            consumed = workflow.__dict__.get("_consumed_entries", set())
            keep = produced - consumed
            self._drop_workflow_entries(workflow, keep=keep)
            self._track_kept(keep)
            # swap under the guard: finish and a racing abort must close
            # the journal exactly once
            with self._flight_guard:
                j, self._journal = self._journal, None
            if j is not None:
                j.close()

    def _track_kept(self, keep: Set[str]) -> None:
        """Bound the post-exit leaf window.  Kept leaves stay hot for
        post-exit reads; across MANY workflows they would accumulate
        forever (one+ per workflow), so the oldest beyond keep_hot_max
        are dropped pool-wide.  Reads of leaves older than the window
        fall back to the durable tier (cache/whiteboard blobs) or raise —
        the deliberate trade vs the reference's serialize-everything-
        to-S3 hot path (see docs/architecture.md)."""
        from collections import OrderedDict

        from lzy_amd.config import get_config

        pool = self.pool
        kept: "OrderedDict[str, None]" = getattr(pool, "kept_hot", None)
        if kept is None:
            kept = pool.kept_hot = OrderedDict()
        for eid in keep:
            kept.pop(eid, None)
            kept[eid] = None
        cap = int(getattr(get_config(), "keep_hot_max", 10000))
        if cap > 0 and len(kept) > cap:
            evict = [kept.popitem(last=False)[0]
                     for _ in range(len(kept) - cap // 2)]
            pool.driver_ctrl.broadcast(
                {"cmd": "drop_entries", "entries": evict}
            )
            METRICS.inc("lzy_kept_evicted", len(evict))

    def abort(self, workflow: "LzyWorkflow") -> None:
        """Stop the graph, then finish.  When a scheduler batch is live
        (another thread is blocked in its barrier), request a mid-flight
        StopGraph and wait for the drain before releasing resources
        (reference: AbortExecution stops graphs first,
        AbortExecution.java:18; StopGraph workflow-service.proto:12-26).
        """
        sched = self._active_sched
        if sched is not None and sched.workflow is workflow:
            sched.request_stop("workflow aborted by client")
            sched._done_evt.wait(timeout=60.0)
        self.finish(workflow)

    def _drop_workflow_entries(self, workflow: "LzyWorkflow",
                               keep: Optional[Set[str]] = None) -> None:
        """Free this workflow's values on every rank (HBM pressure control).

        Persisted blobs (cache hits, whiteboard fields) survive on the
        durable tier; post-exit proxy materialization falls back to them.
        ``keep`` (leaf results) stay hot for post-exit reads — the spill
        tier bounds their HBM footprint.
        """
        keep = keep or set()
        entry_ids = [
            e for e in workflow.snapshot._entries.keys() if e not in keep
        ]
        if not entry_ids:
            return
        pool = self.pool
        pool.driver_ctrl.broadcast({"cmd": "drop_entries", "entries": entry_ids})

    # -- entry materialization on the driver --------------------------------

    def fetch_entry(self, entry_id: str, meta: EntryMeta) -> None:
        """Pull an entry owned by another rank into rank 0's store."""
        pool = self.pool
        owner = next(iter(meta.owners - {0}), None)
        if owner is None:
            return
        xtag = pool.next_seq() % (1 << 30)
        pool.driver_ctrl.send(
            owner,
            {"cmd": "xfer_send_batch",
             "items": [{"entry": entry_id, "dst": 0, "tag": xtag}]},
        )
        pool.driver_ctrl.send(
            0,
            {"cmd": "xfer_recv_batch",
             "items": [{"entry": entry_id, "src": owner,
                        "meta": meta.to_wire(), "tag": xtag}]},
        )
        tag = f"f{pool.next_seq()}"
        pool.driver_ctrl.send(0, {"cmd": "settle", "entries": [entry_id], "tag": tag})
        pool.wait_acks("settled", tag, [0])
        meta.owners.add(0)


class _DriverScheduler:
    """One barrier batch: DAG -> placement -> dispatch -> completion."""

    def __init__(self, pool: GpuPool, workflow: "LzyWorkflow",
                 calls: Sequence["LzyCall"], journal) -> None:
        self.pool = pool
        self.workflow = workflow
        self.calls = {c.id: c for c in calls}
        self.journal = journal
        self.meta: Dict[str, EntryMeta] = getattr(workflow, "_entry_meta", {})
        workflow._entry_meta = self.meta
        # dead ranks stay excluded across batches (a later workflow must
        # not dispatch to a corpse)
        self.outstanding: Dict[int, int] = {
            r: 0 for r in range(pool.world)
            if r not in getattr(pool, "dead_ranks", set())
        }
        self.task_ranks: Dict[str, List[int]] = {}
        self.task_dispatch_ts: Dict[str, float] = {}
        self.gang_pending: Dict[str, Set[int]] = {}
        self.errors: List[BaseException] = []
        self.inflight = 0
        # entries whose transfer to a rank was initiated in this batch:
        # with intra-rank task concurrency, every later consumer task on
        # that rank must also wait for the settle (taskspec.wait_present)
        self.transferred_now: Set[Tuple[int, str]] = set()
        # collectives must start in the same order on every rank.  Gangs
        # with DISJOINT rank sets cannot interleave collectives on any
        # rank, so they run concurrently (reference: per-task LRO
        # concurrency, ExecuteTaskAction.java:44); a gang that cannot
        # get a disjoint slot queues here.  Group creation stays
        # driver-sequenced (ensure_group broadcast + all-rank acks).
        self.deferred_gangs: List[str] = []
        # chain dispatch: a task whose producers are all inflight on ONE
        # rank is dispatched there immediately (worker FIFO + store
        # condition waits give correct ordering) — the driver round-trip
        # leaves the dependency-chain critical path
        self.task_deps: Dict[str, List[str]] = {}
        self.task_children: Dict[str, List[str]] = {}
        # chain-eligibility worklist: a task is (re)examined only when
        # one of its producers gets dispatched — scanning ALL calls per
        # completion was O(n^2) on wide graphs (measured 0.27 -> 0.54
        # ms/op from width 128 to 1024)
        self._chain_candidates: Set[str] = set()
        self.entry_producer: Dict[str, str] = {}
        self.dispatched: Set[str] = set()
        self.dag_completed: Set[str] = set()
        self.done_pending: Dict[str, str] = {}  # tid -> completed-not-dag-acked
        self.chained_waits: Dict[str, Tuple[int, List[str]]] = {}
        from lzy_amd.config import get_config

        # ONE config resolve per batch: get_config() re-fingerprints the
        # LZY_* environment, which costs ~70 us a call — 8+ calls per
        # two-batch DAG showed up in the host profile
        cfg = get_config()
        self._settle_wait = float(getattr(cfg, "settle_wait_s", 120.0))
        self._chain_enabled = bool(getattr(cfg, "chain_dispatch", True))
        # worker-death recovery: per-task re-dispatch budget (reference:
        # scheduler re-allocation after VM death + storage-peer failover)
        self._max_retries = int(getattr(cfg, "task_retries", 1))
        self.retry_budget: Dict[str, int] = {}
        self.retrying: Set[str] = set()
        # mid-flight StopGraph: reason once set; done event lets an
        # aborting thread wait for the drain to finish
        self.stopping: Optional[str] = None
        self._done_evt = threading.Event()
        # (task_id, rank) -> [out_chars, err_chars] already streamed live
        self.streamed_logs: Dict[Tuple[str, int], list] = {}
        # streamed merge-tree plans (channels/treeplan.py): connected
        # components of pair_reduce tasks fold into one chunk-pipelined
        # multi-rank reduction instead of log2(N) transfer-then-combine
        # levels.  One plan active pool-wide (per-pair p2p issue order).
        self.stream_components: Dict[str, int] = {}   # tid -> comp idx
        self.components: Dict[int, dict] = {}
        self.active_plan: Optional[dict] = None
        self.deferred_plans: List[int] = []
        self._stream_chunk = int(getattr(cfg, "stream_chunk_mb", 64)) << 20
        # explicit ipc mode, or automatic when ranks outnumber GPUs:
        # RCCL cannot build a comm then, and the host-staged fallback is
        # ~19x slower than hipIpc zero-copy for device tensors on one
        # node (profiles/bench_history.md)
        explicit_ipc = cfg.channel_transport == "ipc"
        agent_tr = getattr(pool.agent, "transport", None)
        self._ipc_mode = explicit_ipc or (
            torch.cuda.is_available()
            and agent_tr is not None
            and not agent_tr._cuda_p2p
        )
        # plans stay on under AUTO-ipc (oversubscribed harness: the plan
        # executor host-stages CUDA chunks through gloo), so the 1-GPU
        # rehearsal exercises the same schedule the 8-GPU RCCL run uses;
        # an EXPLICIT ipc transport choice disables them
        self._stream_merge_on = (
            pool.world > 1
            and bool(getattr(cfg, "stream_merge", True))
            and not explicit_ipc
        )

    def _discover_components(self) -> None:
        """Find foldable pair_reduce components (called from run() once
        task_deps is built)."""
        if not self._stream_merge_on:
            return
        from lzy_amd.channels.treeplan import find_components

        for ci, member_order in enumerate(find_components(self.calls)):
            member_set = set(member_order)
            entry_members = {
                t for t in member_order
                if not any(d in member_set for d in self.task_deps.get(t, ()))
            }
            self.components[ci] = {
                "order": member_order,
                "members": member_set,
                "entry_members": entry_members,
                "ready": set(),
                "broken": False,
                "launched": False,
            }
            for t in member_order:
                self.stream_components[t] = ci

    # -- metadata helpers ---------------------------------------------------

    def _meta_for_driver_entry(self, eid: str) -> EntryMeta:
        m = self.meta.get(eid)
        if m is None:
            value = self.pool.agent.store.get(eid)
            m = describe_value(eid, value)
            if m.kind == KIND_BYTES:
                data = pickle_value(value)
                self.pool.agent.store.pickled[eid] = data
                m.nbytes = len(data)
            m.owners = {0}
            self.meta[eid] = m
        return m

    # -- main loop ----------------------------------------------------------

    def run(self) -> None:
        t0 = time.perf_counter()
        producer: Dict[str, str] = {}
        for c in self.calls.values():
            for eid in c.entry_ids:
                producer[eid] = c.id

        dag = Dag()
        for c in self.calls.values():
            deps = sorted({
                producer[eid]
                for eid in c.input_entry_ids()
                if eid in producer and producer[eid] != c.id
            })
            dag.add_task(c.id, deps)
            self.task_deps[c.id] = deps
            for d in deps:
                self.task_children.setdefault(d, []).append(c.id)
        self.entry_producer = producer
        dag.seal()
        self._discover_components()
        METRICS.observe("lzy_graph_build", time.perf_counter() - t0)

        for tid in dag.take_ready():
            self._dispatch(tid)
        self._try_chain()

        failed_tasks: Set[str] = set()
        self._failed_tasks = failed_tasks
        try:
            self._event_loop(dag, failed_tasks)
        except (KeyboardInterrupt, SystemExit):
            # Ctrl-C mid-barrier: stop the graph (cancel queued tasks on
            # every rank, poison blocked waits) and drain what is truly
            # running, bounded — then let the interrupt propagate
            self._initiate_stop(dag, "interrupted (Ctrl-C)")
            try:
                self._event_loop(
                    dag, failed_tasks, deadline=time.monotonic() + 30.0
                )
            except BaseException:  # noqa: BLE001 - interrupt wins
                pass
            raise
        finally:
            self._done_evt.set()

        if self.stopping is not None:
            from lzy_amd.exceptions import WorkflowAbortedError

            raise WorkflowAbortedError(
                f"graph stopped: {self.stopping}"
            )
        if self.errors:
            raise self.errors[0]

    def request_stop(self, reason: str) -> None:
        """Thread-safe mid-flight StopGraph (reference: StopGraph RPC,
        workflow-service.proto:12-26; AbortExecution.java:18).  Wakes the
        event loop; it cancels everything not yet running and drains."""
        self.pool.events.put(
            (-1, {"ev": "stop", "reason": reason, "sched": id(self)})
        )

    def _initiate_stop(self, dag, reason: str) -> None:
        if self.stopping is not None:
            return
        self.stopping = reason
        pool = self.pool
        self.deferred_gangs = []
        for cid in self.deferred_plans:
            for m in self.components[cid]["order"]:
                self.journal.record(m, "cancelled", reason)
        self.deferred_plans = []
        for tid in self.calls:
            if tid not in self.dispatched:
                self.journal.record(tid, "cancelled", reason)
        live = list(self.task_dispatch_ts)
        if live:
            pool.driver_ctrl.broadcast(
                {"cmd": "cancel_tasks", "ids": live, "reason": reason}
            )
        # blocked settles (chained consumers waiting on producers that
        # will now never run) must fail fast, not time out
        for tid, (rank, eids) in list(self.chained_waits.items()):
            try:
                pool.driver_ctrl.send(rank, {
                    "cmd": "poison", "entries": eids, "reason": reason,
                })
            except (OSError, KeyError, BrokenPipeError):
                pass
        _LOG.warning("graph stop initiated: %s (%d tasks inflight)",
                     reason, self.inflight)

    def _event_loop(self, dag, failed_tasks: Set[str],
                    deadline: Optional[float] = None) -> None:
        pool = self.pool
        while self.inflight > 0:
            if deadline is None:
                rank, msg = pool.events.get()
            else:
                try:
                    rank, msg = pool.events.get(
                        timeout=max(0.05, deadline - time.monotonic())
                    )
                except queue.Empty:
                    _LOG.warning(
                        "stop drain abandoned with %d tasks inflight",
                        self.inflight,
                    )
                    return
            ev = msg.get("ev")
            if ev in ("task_done", "task_failed"):
                tid = msg["result"].task_id
                if tid not in self.calls or tid not in self.task_dispatch_ts:
                    # stale event: a prior batch's leftover, or a task
                    # already terminally accounted (agent_error /
                    # worker_lost) — must not touch inflight again
                    continue
            if ev == "task_done":
                result: TaskResult = msg["result"]
                finished = self._on_done(rank, result)
                if finished and result.task_id not in failed_tasks:
                    # chained completions can outrace their parents'
                    # events; feed the DAG in dependency order
                    self.done_pending[result.task_id] = result.task_id
                    progressed = True
                    while progressed:
                        progressed = False
                        for t in list(self.done_pending):
                            if all(d in self.dag_completed
                                   for d in self.task_deps.get(t, ())):
                                self.done_pending.pop(t)
                                self.dag_completed.add(t)
                                for nxt in dag.complete(t):
                                    if nxt not in self.dispatched:
                                        self._dispatch(nxt)
                                progressed = True
                    self._try_chain()
            elif ev == "task_failed":
                result = msg["result"]
                self._on_failed(rank, result)
                if result.task_id not in failed_tasks:
                    failed_tasks.add(result.task_id)
                    for ct in dag.fail(result.task_id):
                        if ct in self.dispatched:
                            # chain-dispatched dependent already inflight:
                            # poison its pending inputs so its settle
                            # fails now instead of timing out
                            failed_tasks.add(ct)
                            self._poison_chained(ct, result.task_id)
                        else:
                            self.journal.record(ct, "cancelled")
            elif ev == "agent_error":
                tid = msg.get("task_id")
                if (
                    tid is not None
                    and tid in self.calls
                    and tid not in self.task_dispatch_ts
                ):
                    # stale: the task was already terminally accounted
                    # (or RETRIED) — its late settle-timeout error must
                    # not fail a workflow that recovered
                    continue
                if tid is None or tid not in self.task_dispatch_ts:
                    # serve-loop failure on a non-task command (transfer
                    # staging etc.): the waiting task surfaces its own
                    # failure later — decrementing inflight here would
                    # underflow and leak its completion event into the
                    # next batch's scheduler
                    self.errors.append(
                        LzyExecutionError(f"agent rank {rank}: {msg['error']}")
                    )
                    continue
                if rank in self.outstanding:
                    self.outstanding[rank] -= 1
                if self._retry_after_agent_error(rank, tid, msg):
                    continue
                self.errors.append(
                    LzyExecutionError(f"agent rank {rank}: {msg['error']}")
                )
                if tid not in failed_tasks:
                    failed_tasks.add(tid)
                    self.journal.record(tid, "failed", msg["error"])
                    for ct in dag.fail(tid):
                        if ct in self.dispatched:
                            failed_tasks.add(ct)
                            self._poison_chained(ct, tid)
                        else:
                            self.journal.record(ct, "cancelled")
                gang = self.gang_pending.get(tid)
                if gang is not None:
                    gang.discard(rank)
                    if gang:
                        continue  # other gang members still must report
                    self.gang_pending.pop(tid, None)
                    self._release_gang(tid)
                self.inflight -= 1
                self.task_dispatch_ts.pop(tid, None)
                self.chained_waits.pop(tid, None)
            elif ev in ("plan_done", "plan_failed"):
                self._on_plan_event(rank, msg, dag, failed_tasks)
            elif ev == "log_chunk":
                self._print_log_chunk(msg)
            elif ev == "task_cancelled":
                tid = msg["task_id"]
                if tid not in self.task_dispatch_ts:
                    continue
                if rank in self.outstanding:
                    self.outstanding[rank] -= 1
                gang = self.gang_pending.get(tid)
                if gang is not None:
                    gang.discard(rank)
                    if gang:
                        continue
                    self.gang_pending.pop(tid, None)
                    self._release_gang(tid)
                self.inflight -= 1
                self.task_dispatch_ts.pop(tid, None)
                self.chained_waits.pop(tid, None)
                self.journal.record(tid, "cancelled", self.stopping or "")
            elif ev == "xfer_failed":
                # a source rank could not provide an entry it was asked
                # to send: purge the target's claimed ownership and fail
                # (or retry, via the agent_error machinery) every task on
                # the destination waiting for it
                eid, dst = msg["entry"], msg["dst"]
                m = self.meta.get(eid)
                if m is not None:
                    m.owners.discard(dst)
                self.transferred_now.discard((dst, eid))
                for tid in list(self.task_dispatch_ts):
                    call = self.calls.get(tid)
                    if (
                        call is None
                        or dst not in self.task_ranks.get(tid, ())
                        or eid not in call.input_entry_ids()
                    ):
                        continue
                    pool.events.put((dst, {
                        "ev": "agent_error", "task_id": tid,
                        "error": (
                            f"transfer of {eid} from rank {rank} failed: "
                            f"{msg.get('error', '')}"
                        ),
                    }))
            elif ev == "stop":
                if msg.get("sched") == id(self):
                    self._initiate_stop(dag, msg.get("reason", "stopped"))
                # a foreign scheduler's stop (stale from a prior batch)
                # is dropped
            elif ev == "worker_lost":
                self._on_worker_lost(rank, dag, failed_tasks)
            # barrier_done etc. are routed via acks, not here

    # -- dispatch ------------------------------------------------------------

    def _dispatch(self, task_id: str, chain_rank: Optional[int] = None,
                  pending_local: Optional[Set[str]] = None) -> None:
        if self.stopping is not None:
            self.journal.record(task_id, "cancelled", self.stopping)
            return
        cid = self.stream_components.get(task_id)
        if cid is not None and chain_rank is None:
            comp = self.components[cid]
            if not comp["broken"]:
                self._plan_member_ready(task_id, comp, cid)
                return
        call = self.calls[task_id]
        gpu_count = call.env.provisioning.effective_gpu_count
        pool = self.pool

        if gpu_count > pool.world:
            self.errors.append(BadProvisioningError(
                f"op {call.callable_name} needs gpu_count={gpu_count}, pool has "
                f"{pool.world} ranks"
            ))
            return

        dispatch_t0 = time.perf_counter()
        if chain_rank is not None:
            ranks = [chain_rank]
            gang = None
        elif gpu_count > 1:
            busy_gang_ranks: Set[int] = set()
            for t in self.gang_pending:
                busy_gang_ranks.update(self.task_ranks.get(t, ()))
            ranks = self._pick_gang(gpu_count, exclude=busy_gang_ranks)
            if ranks is None:
                self.deferred_gangs.append(task_id)
                return
            try:
                tag = pool.ensure_group(ranks)
            except RuntimeError as e:
                # post-death pools cannot create NEW groups (collective
                # over all ranks): fail the gang op with a typed error
                # instead of crashing the scheduler loop
                self.errors.append(BadProvisioningError(str(e)))
                self.journal.record(task_id, "failed", str(e))
                return
            gang = {"ranks": ranks, "tag": tag}
        else:
            ranks = [self._pick_rank(call)]
            gang = None

        self.dispatched.add(task_id)
        self.task_ranks[task_id] = ranks
        if gang is not None:
            self.gang_pending[task_id] = set(ranks)

        specs_inline: Dict[str, bytes] = {}
        wait_entries_per_rank: Dict[int, List[str]] = {r: [] for r in ranks}
        sends_by_owner: Dict[int, List[dict]] = {}
        recvs_by_rank: Dict[int, List[dict]] = {}
        ipc_exports: Dict[int, List[str]] = {}
        ipc_imports: List[Tuple[int, str]] = []

        for eid in call.input_entry_ids():
            if pending_local and eid in pending_local:
                # produced by an inflight task on this same rank: no
                # transfer, the worker's settle waits on the store
                wait_entries_per_rank[ranks[0]].append(eid)
                continue
            meta = self.meta.get(eid)
            if meta is None:
                # driver-captured arg (or earlier-batch result on rank 0)
                meta = self._meta_for_driver_entry(eid)
            if (
                meta.kind == KIND_BYTES
                and meta.nbytes <= INLINE_LIMIT
                and 0 in meta.owners
                # owners gains rank 0 when a transfer is merely INITIATED;
                # inline only when the value actually landed here
                and self.pool.agent.store.has(eid)
            ):
                specs_inline[eid] = self.pool.agent.store.pickled.get(
                    eid
                ) or pickle_value(self.pool.agent.store.get(eid))
                continue
            from lzy_amd.channels.transport import KIND_TENSOR as _KT

            use_ipc = (
                self._ipc_mode
                and meta.kind == _KT
                and meta.device_type == "cuda"
            )
            for r in ranks:
                if r not in meta.owners:
                    # source must be a CONFIRMED owner — ranks in
                    # transferred_now were only targeted, their copy may
                    # not have settled yet
                    confirmed = [
                        o for o in meta.owners
                        if (o, eid) not in self.transferred_now
                    ]
                    pickable = confirmed or sorted(meta.owners)
                    owner = 0 if 0 in pickable else pickable[0]
                    if use_ipc:
                        # zero-copy path: map the producer's HBM allocation
                        # in the consumer (hipIpc; same GPU = no copy,
                        # cross GPU = one xGMI DMA on import).  Handle
                        # exports are batched per owner: one ack round
                        # trip per task, not per entry.
                        if meta.ipc_handle is None:
                            ipc_exports.setdefault(owner, []).append(eid)
                        ipc_imports.append((r, eid))
                        METRICS.inc("lzy_transfers_ipc")
                    else:
                        # unique per-transfer tag: gloo matches (pair,
                        # tag, order), so a FAILED transfer's stale
                        # posted recvs can never swallow a later
                        # transfer's chunks on the same pair
                        xtag = pool.next_seq() % (1 << 30)
                        sends_by_owner.setdefault(owner, []).append(
                            {"entry": eid, "dst": r, "tag": xtag}
                        )
                        recvs_by_rank.setdefault(r, []).append(
                            {"entry": eid, "src": owner,
                             "meta": meta.to_wire(), "tag": xtag}
                        )
                        METRICS.inc("lzy_transfers")
                    wait_entries_per_rank[r].append(eid)
                    self.transferred_now.add((r, eid))
                    meta.owners.add(r)
                    METRICS.inc("lzy_transfer_bytes", meta.nbytes)
                elif (r, eid) in self.transferred_now:
                    # transfer already initiated for an earlier task this
                    # batch; this task must still wait for its settle
                    wait_entries_per_rank[r].append(eid)

        # batched hipIpc handle exchange: one export request per owner,
        # then the imports
        if ipc_exports:
            tags = {}
            for owner, eids in ipc_exports.items():
                tag = f"ipc{pool.next_seq()}"
                tags[owner] = (tag, eids)
                pool.driver_ctrl.send(
                    owner, {"cmd": "ipc_export_batch", "entries": eids,
                            "tag": tag}
                )
            for owner, (tag, eids) in tags.items():
                payloads = pool.wait_acks("ack", tag, [owner])
                handles = payloads[owner]
                for eid in eids:
                    self.meta[eid].ipc_handle = handles[eid]
        for r, eid in ipc_imports:
            pool.driver_ctrl.send(
                r, {"cmd": "ipc_import", "entry": eid,
                    "data": self.meta[eid].ipc_handle}
            )

        # one grouped send/recv command per rank for this task's transfers:
        # the worker issues the whole group through ONE batch_isend_irecv
        # (ncclGroupStart/End), so transfers with distinct peers progress
        # in parallel over their own xGMI links
        for owner, items in sends_by_owner.items():
            pool.driver_ctrl.send(owner, {"cmd": "xfer_send_batch", "items": items})
        for r, items in recvs_by_rank.items():
            pool.driver_ctrl.send(r, {"cmd": "xfer_recv_batch", "items": items})

        func_bytes = _func_bytes(call.signature.func)
        snap = self.workflow.snapshot
        for i, r in enumerate(ranks):
            spec = TaskSpec(
                task_id=task_id,
                name=call.callable_name,
                func_bytes=func_bytes,
                arg_entries=list(call.arg_entry_ids),
                kwarg_entries=dict(call.kwarg_entry_ids),
                output_entries=[
                    (eid, snap.get_entry(eid).storage_uri) for eid in call.entry_ids
                ],
                exception_entry=call.exception_id,
                env_vars=dict(call.env.env_variables),
                execution_id=self.workflow.execution_id,
                cache=call.cache,
                version=call.version,
                storage_root=self.workflow.owner.storage_uri,
                inline_values=specs_inline,
                wait_entries=wait_entries_per_rank[r],
                gang={**gang, "gang_rank": i} if gang is not None else None,
            )
            pool.driver_ctrl.send(r, {"cmd": "task", "spec": spec})
            self.outstanding[r] += 1
        self.inflight += 1
        if chain_rank is not None and pending_local:
            self.chained_waits[task_id] = (chain_rank, sorted(pending_local))
        self.task_dispatch_ts[task_id] = dispatch_t0
        self._chain_candidates.update(self.task_children.get(task_id, ()))
        self.journal.record(task_id, "scheduled", call.callable_name)
        METRICS.observe("lzy_dispatch", time.perf_counter() - dispatch_t0)

    def _pick_rank(self, call: "LzyCall") -> int:
        # data affinity: the rank already holding the most input bytes;
        # ties broken by load
        byrank: Dict[int, int] = {}
        for eid in call.input_entry_ids():
            meta = self.meta.get(eid)
            if meta is None or meta.kind != KIND_TENSOR:
                continue
            for r in meta.owners:
                byrank[r] = byrank.get(r, 0) + meta.nbytes
        if byrank:
            best = max(byrank.items(), key=lambda kv: (kv[1], -self.outstanding[kv[0]]))
            return best[0]
        # no data affinity: least-loaded, ties broken round-robin — a
        # fixed min-rank tie-break collapses fast-completing fan-outs
        # onto rank 0 (completions race dispatch; single-task barriers
        # from eager materialization would otherwise all see a fresh
        # batch).  The rotation lives on the pool so it survives batches.
        lo = min(self.outstanding.values())
        cands = sorted(r for r, o in self.outstanding.items() if o == lo)
        self.pool.rr_counter = getattr(self.pool, "rr_counter", -1) + 1
        return cands[self.pool.rr_counter % len(cands)]

    def _pick_gang(self, k: int,
                   exclude: Optional[Set[int]] = None) -> Optional[List[int]]:
        """Least-loaded k ranks disjoint from every inflight gang;
        None when no disjoint slot exists right now."""
        exclude = exclude or set()
        cands = [r for r in self.outstanding if r not in exclude]
        if len(cands) < k:
            return None
        ranks = sorted(cands, key=lambda r: (self.outstanding[r], r))[:k]
        return sorted(ranks)

    # -- completion ----------------------------------------------------------

    def _print_log_chunk(self, msg: dict) -> None:
        """A running remote op's incremental std-logs (reference:
        ReadStdSlots server-stream fed live from Kafka,
        KafkaLogsListeners.java:35): print now, remember how much was
        streamed so the completion-time tail prints only the rest."""
        name = msg["name"]
        out, err = msg.get("out", ""), msg.get("err", "")
        for line in out.splitlines():
            print(f"[LZY-{name}] {line}", flush=True)
        for line in err.splitlines():
            print(f"[LZY-{name}] {line}", file=sys.stderr, flush=True)
        shipped = self.streamed_logs.setdefault(
            (msg["task_id"], msg["rank"]), [0, 0]
        )
        shipped[0] += len(out)
        shipped[1] += len(err)

    def _tail_logs(self, rank: int, result: TaskResult) -> None:
        """Print a remote rank's captured op logs on the client console
        (reference: ReadStdSlots live tail, runtime.py:283-301).  Lines
        already streamed live via log_chunk events are skipped."""
        if rank == 0:
            return  # echoed live on the shared console already
        name = self.calls[result.task_id].callable_name
        so, se = self.streamed_logs.pop((result.task_id, rank), (0, 0))
        if result.logs_out[so:]:
            for line in result.logs_out[so:].splitlines():
                print(f"[LZY-{name}] {line}", flush=True)
        if result.logs_err[se:]:
            for line in result.logs_err[se:].splitlines():
                print(f"[LZY-{name}] {line}", file=sys.stderr, flush=True)

    def _on_done(self, rank: int, result: TaskResult) -> bool:
        """Returns True when the task fully completed (all gang members)."""
        self.outstanding[rank] -= 1
        self._tail_logs(rank, result)
        call = self.calls[result.task_id]
        gang = self.gang_pending.get(result.task_id)
        primary = self.task_ranks[result.task_id][0]
        if rank == primary:
            self._record_outputs(call, rank, result)
        if gang is not None:
            gang.discard(rank)
            if gang:
                return False  # wait for the rest of the gang
            self.gang_pending.pop(result.task_id, None)
            self._release_gang(result.task_id)
        self.inflight -= 1
        self.chained_waits.pop(result.task_id, None)
        self.journal.record(result.task_id, "done")
        ts = self.task_dispatch_ts.pop(result.task_id, None)
        if ts is not None:
            # full round trip minus the op itself = framework overhead/task
            METRICS.observe(
                "lzy_task_overhead",
                max(0.0, time.perf_counter() - ts - result.elapsed_s),
            )
        if result.cached:
            METRICS.inc("lzy_cache_hits_pool")
        return True

    def _poison_chained(self, child_tid: str, failed_tid: str) -> None:
        hit = self.chained_waits.get(child_tid)
        if hit is None:
            return
        rank, eids = hit
        name = self.calls[failed_tid].callable_name
        try:
            self.pool.driver_ctrl.send(rank, {
                "cmd": "poison", "entries": eids,
                "reason": f"producer op {name} failed",
            })
        except (OSError, KeyError, BrokenPipeError):
            # the chained child shared the dead producer's rank: its
            # tasks were already failed by the worker_lost sweep
            pass

    # -- streamed merge-tree plans ------------------------------------------

    def _plan_member_ready(self, tid: str, comp: dict, cid: int) -> None:
        """A component member became DAG-ready.  Members are held back
        until every entry member is ready (interior members only become
        ready through the plan itself), then the whole component launches
        as one streamed plan — or breaks apart to op-by-op execution if
        the leaves don't verify (shape/dtype mismatch etc.)."""
        self.dispatched.add(tid)
        comp["ready"].add(tid)
        if not comp["entry_members"] <= comp["ready"] or comp["launched"]:
            return
        if self.active_plan is not None:
            # one plan active pool-wide: per-(src,dst) p2p issue order on
            # pg_stream must stay sequential
            if cid not in self.deferred_plans:
                self.deferred_plans.append(cid)
            return
        self._launch_plan(cid, comp)

    def _break_component(self, comp: dict) -> None:
        comp["broken"] = True
        ready = sorted(comp["ready"])
        comp["ready"] = set()
        for m in ready:
            self.dispatched.discard(m)
        for m in ready:
            if m not in self.dispatched:
                self._dispatch(m)
        self._try_chain()

    def _launch_plan(self, cid: int, comp: dict) -> None:
        from lzy_amd.channels.treeplan import build_plan

        pool = self.pool

        def meta_of(eid: str):
            m = self.meta.get(eid)
            if m is None:
                try:
                    m = self._meta_for_driver_entry(eid)
                except KeyError:
                    return None
            # only CONFIRMED owners may stream (cf. transfer sourcing)
            confirmed = {
                o for o in m.owners
                if (o, eid) not in self.transferred_now and o in self.outstanding
            }
            if not confirmed:
                return None
            import copy as _copy

            m2 = _copy.copy(m)
            m2.owners = confirmed
            return m2

        agent_tr = getattr(pool.agent, "transport", None)
        plan = build_plan(
            f"plan-{cid}-{pool.next_seq()}",
            comp["order"], self.calls, meta_of,
            chunk_bytes=self._stream_chunk,
            cuda_p2p=bool(agent_tr is not None and agent_tr._cuda_p2p),
        )
        if plan is None:
            _LOG.info("stream component %d not foldable — op-by-op", cid)
            self._break_component(comp)
            return
        comp["launched"] = True
        steps_by_rank = plan.pop("steps_by_rank")
        t0 = time.perf_counter()
        self.active_plan = {
            "cid": cid,
            "plan_id": plan["plan_id"],
            "pending": set(plan["participants"]),
            "node_rank": dict(plan["node_rank"]),
            "outputs": [],
            "failed": [],
            "t0": t0,
        }
        wait_s = self._settle_wait
        for r in plan["participants"]:
            pool.driver_ctrl.send(r, {
                "cmd": "stream_plan", "plan": plan,
                "steps": steps_by_rank[r], "wait_timeout": wait_s,
            })
            if r in self.outstanding:
                self.outstanding[r] += 1
        self.inflight += 1
        for m in comp["order"]:
            self.journal.record(m, "scheduled", self.calls[m].callable_name)
        # account the cross-rank edges as transfers (bench/SCALE telemetry)
        n_edges = sum(
            1 for sts in steps_by_rank.values() for st in sts
            if st["op"] == "leaf_send"
        ) + sum(
            len(st.get("send_to", ())) for sts in steps_by_rank.values()
            for st in sts
        )
        elem = torch.empty(0, dtype=getattr(torch, plan["dtype"])).element_size()
        METRICS.inc("lzy_stream_plans")
        METRICS.inc("lzy_transfers", n_edges)
        METRICS.inc("lzy_transfer_bytes", n_edges * plan["numel"] * elem)
        _LOG.info(
            "stream plan %s: %d nodes on ranks %s, %d cross edges",
            plan["plan_id"], len(comp["order"]), plan["participants"], n_edges,
        )

    def _on_plan_event(self, rank: int, msg: dict, dag,
                       failed_tasks: Set[str]) -> None:
        ap = self.active_plan
        if ap is None or msg.get("plan_id") != ap["plan_id"]:
            return  # stale
        ev = msg["ev"]
        ap["pending"].discard(rank)
        if rank in self.outstanding:
            self.outstanding[rank] -= 1
        if ev == "plan_done":
            ap["outputs"].extend(msg.get("outputs", ()))
        else:
            ap["failed"].append((rank, msg.get("error", "plan failed")))
        if ap["pending"]:
            return
        # all participants reported: resolve the whole component
        self.active_plan = None
        self.inflight -= 1
        comp = self.components[ap["cid"]]
        if not ap["failed"]:
            METRICS.observe("lzy_stream_plan_s", time.perf_counter() - ap["t0"])
            for w in ap["outputs"]:
                ps = w.get("plan_stats")
                if ps:  # per-rank recv-wait (SCALE overlap evidence)
                    METRICS.observe(
                        "lzy_plan_recv_wait_s", float(ps.get("recv_wait_s", 0.0))
                    )
            by_task = {w["task"]: w for w in ap["outputs"]}
            for m in comp["order"]:
                wire = by_task.get(m)
                if wire is not None:
                    meta = EntryMeta.from_wire(wire)
                    # the node's compute rank (from the plan builder)
                    # owns the materialized accumulator
                    meta.owners = {ap["node_rank"].get(m, 0)}
                    self.meta[meta.entry_id] = meta
                self.journal.record(m, "done")
                self.dag_completed.add(m)
                for nxt in dag.complete(m):
                    if nxt not in self.dispatched:
                        self._dispatch(nxt)
            self._try_chain()
        else:
            rank0, err = ap["failed"][0]
            for m in comp["order"]:
                self.errors.append(LzyExecutionError(
                    f"streamed merge plan failed on rank {rank0}: {err}",
                    task_id=m,
                ))
                self.journal.record(m, "failed", err)
                if m not in failed_tasks:
                    failed_tasks.add(m)
                    for ct in dag.fail(m):
                        if ct not in comp["members"] and ct not in self.dispatched:
                            self.journal.record(ct, "cancelled")
        # a deferred plan may launch now
        while self.deferred_plans and self.active_plan is None:
            nxt_cid = self.deferred_plans.pop(0)
            nxt_comp = self.components[nxt_cid]
            if not nxt_comp["launched"] and not nxt_comp["broken"]:
                self._launch_plan(nxt_cid, nxt_comp)

    def _retry_after_agent_error(self, rank: int, tid: str, msg: dict) -> bool:
        """A task's settle failed (typically: its input's source rank
        died mid-transfer, poisoning or timing out the wait).  If the
        retry budget allows and every input is still recoverable, account
        the attempt as terminated and re-dispatch instead of failing the
        workflow (the same failover worker death itself gets)."""
        call = self.calls.get(tid)
        if (
            call is None
            or self.retry_budget.get(tid, self._max_retries) <= 0
            or call.env.provisioning.effective_gpu_count > 1
            or tid in self.gang_pending
            or self.stopping is not None
            or not self.outstanding
        ):
            _LOG.warning("agent-error retry declined for %s (budget=%s)",
                         getattr(call, "callable_name", tid),
                         self.retry_budget.get(tid, self._max_retries))
            return False
        # recoverability uses the same machinery as worker death; no
        # concurrent sweep is running, so retrying={tid} only
        self.retrying = {tid}
        ok = self._recover_inputs(call, prefer_reliable=True)
        self.retrying = set()
        if not ok:
            _LOG.warning("agent-error retry: inputs unrecoverable for %s",
                         call.callable_name)
            return False
        self.retry_budget[tid] = self.retry_budget.get(tid, self._max_retries) - 1
        self.inflight -= 1
        self.task_dispatch_ts.pop(tid, None)
        self.chained_waits.pop(tid, None)
        self.task_ranks.pop(tid, None)
        self.dispatched.discard(tid)
        self.journal.record(tid, "retry", f"agent error on rank {rank}")
        METRICS.inc("lzy_task_retries")
        _LOG.warning(
            "task %s failed on rank %d (%s); re-dispatching",
            call.callable_name, rank, msg.get("error", ""),
        )
        if all(d in self.dag_completed for d in self.task_deps.get(tid, ())):
            self._dispatch(tid)
        return True

    def _reroot_on_driver(self, eid: str, meta, snap, store) -> bool:
        """Make the driver a confirmed owner of ``eid`` from its own
        store or the durable tier; True on success."""
        if store.has(eid):
            meta.owners.add(0)
            return True
        try:
            entry = snap.get_entry(eid)
        except KeyError:
            _LOG.warning("reroot: no snapshot entry for %s", eid)
            return False
        if not snap.storage.blob_exists(entry.storage_uri):
            _LOG.warning("reroot: no durable blob at %s", entry.storage_uri)
            return False
        value = snap.load(eid)  # lands in the rank-0 store (shared dict)
        fresh = describe_value(eid, value)
        fresh.owners = {0}
        if fresh.kind == KIND_BYTES:
            data = pickle_value(value)
            store.pickled[eid] = data
            fresh.nbytes = len(data)
        self.meta[eid] = fresh
        return True

    # -- worker death --------------------------------------------------------

    def _on_worker_lost(self, rank: int, dag, failed_tasks: Set[str]) -> None:
        """A worker died (its control socket closed).  Exclude it from
        placement, purge its data ownership, and re-dispatch its inflight
        tasks onto surviving ranks when their inputs are recoverable
        (reference failover: consumers re-pointed at the channel's storage
        peer + scheduler re-allocation, SlotsService.java:191-240);
        unrecoverable tasks fail the workflow as before.

        Not recovered: gang tasks (their RCCL subgroup spans the dead
        rank; surviving members' collectives cannot complete) and tasks
        whose only input copy died with the rank and has no durable blob.
        """
        self.outstanding.pop(rank, None)
        # an active streamed plan touching the dead rank cannot complete:
        # fail its members now (survivors' recv timeouts surface later as
        # stale plan_failed events and are dropped)
        ap = self.active_plan
        if ap is not None and (
            rank in ap["pending"] or rank in ap["node_rank"].values()
        ):
            self.active_plan = None
            self.inflight -= 1
            # survivors' plan_failed events arrive later as stale and are
            # dropped — release their outstanding slots here
            for r in ap["pending"]:
                if r != rank and r in self.outstanding:
                    self.outstanding[r] -= 1
            comp = self.components[ap["cid"]]
            for m in comp["order"]:
                self.errors.append(LzyExecutionError(
                    f"worker rank {rank} died during streamed merge plan",
                    task_id=m,
                ))
                self.journal.record(m, "failed", f"worker {rank} lost")
                if m not in failed_tasks:
                    failed_tasks.add(m)
                    for ct in dag.fail(m):
                        if ct not in comp["members"] and ct not in self.dispatched:
                            self.journal.record(ct, "cancelled")
        for m in self.meta.values():
            if rank in m.owners:
                m.owners.discard(rank)
                # the cached handle may map the dead process's allocation
                m.ipc_handle = None
        self.transferred_now = {
            (r, e) for (r, e) in self.transferred_now if r != rank
        }
        dead = [
            tid for tid, rks in self.task_ranks.items()
            if rank in rks and tid in self.task_dispatch_ts
        ]
        # retry eligibility, then input recovery to a fixpoint: a chained
        # child's lost input is fine iff its producer re-runs in this
        # same sweep — a demoted producer demotes the child too
        self.retrying = {
            tid for tid in dead
            if self.outstanding
            and self.retry_budget.get(tid, self._max_retries) > 0
            and self.calls[tid].env.provisioning.effective_gpu_count <= 1
        }
        changed = True
        while changed:
            changed = False
            for tid in list(self.retrying):
                if not self._recover_inputs(self.calls[tid]):
                    self.retrying.discard(tid)
                    changed = True
        for tid in dead:
            self.task_dispatch_ts.pop(tid, None)
            self.chained_waits.pop(tid, None)
            gang = self.gang_pending.pop(tid, None)
            if gang is not None:
                self._release_gang(tid)
            self.inflight -= 1
            self.task_ranks.pop(tid, None)
            if tid in self.retrying:
                self.retry_budget[tid] = (
                    self.retry_budget.get(tid, self._max_retries) - 1
                )
                self.dispatched.discard(tid)
                self.journal.record(tid, "retry", f"worker {rank} lost")
                METRICS.inc("lzy_task_retries")
                _LOG.warning(
                    "worker rank %d died; re-dispatching %s",
                    rank, self.calls[tid].callable_name,
                )
            else:
                self.errors.append(LzyExecutionError(
                    f"worker rank {rank} died while running "
                    f"{self.calls[tid].callable_name}", task_id=tid,
                ))
                self.journal.record(tid, "failed", f"worker {rank} lost")
                if tid not in failed_tasks:
                    failed_tasks.add(tid)
                    for ct in dag.fail(tid):
                        if ct in self.dispatched:
                            # chained dependent inflight on a LIVE rank:
                            # poison its waits now
                            failed_tasks.add(ct)
                            self._poison_chained(ct, tid)
                        else:
                            self.journal.record(ct, "cancelled")
        # re-dispatch retried tasks whose deps are already satisfied; the
        # rest re-dispatch when their (also-retried) producers complete,
        # through the run loop's normal dag.complete path
        for tid in sorted(self.retrying):
            if tid in self.dispatched:
                continue
            if all(d in self.dag_completed for d in self.task_deps.get(tid, ())):
                self._dispatch(tid)
        self.retrying = set()

    def _recover_inputs(self, call: "LzyCall",
                        prefer_reliable: bool = False) -> bool:
        """Check every input of a to-be-retried task is reachable from a
        surviving rank — restoring driver ownership from the durable tier
        where needed.  Returns False when any input is gone for good.

        ``prefer_reliable`` (agent-error retries): re-root the input on
        the driver/durable tier even when a remote owner LOOKS alive —
        the failure may have raced ahead of that owner's death event, and
        a retry that trusts it just burns the budget."""
        snap = self.workflow.snapshot
        store = self.pool.agent.store
        for eid in call.input_entry_ids():
            prod = self.entry_producer.get(eid)
            if prod is not None and prod in self.retrying:
                continue  # will be re-produced by the retried producer
            meta = self.meta.get(eid)
            if meta is None:
                # driver-captured arg: lives in the rank-0 store/snapshot
                if store.has(eid) or snap.has_value(eid):
                    continue
                return False
            if prefer_reliable and self._reroot_on_driver(eid, meta, snap, store):
                continue
            # only CONFIRMED live owners count: a rank in owners whose
            # copy was merely INITIATED (transferred_now) may never have
            # received it — e.g. the source died mid-send
            if any(
                o in self.outstanding
                and (o, eid) not in self.transferred_now
                for o in meta.owners
            ):
                continue
            if store.has(eid):
                meta.owners.add(0)
                continue
            # durable tier: cache blobs / persisted entries reload on the
            # driver, which becomes the new owner
            try:
                entry = snap.get_entry(eid)
            except KeyError:
                return False
            if not snap.storage.blob_exists(entry.storage_uri):
                return False
            value = snap.load(eid)  # lands in the rank-0 store (shared dict)
            fresh = describe_value(eid, value)
            fresh.owners = {0}
            if fresh.kind == KIND_BYTES:
                data = pickle_value(value)
                store.pickled[eid] = data
                fresh.nbytes = len(data)
            self.meta[eid] = fresh
        return True

    def _try_chain(self) -> None:
        """Dispatch tasks whose producers are all inflight/complete on one
        rank, without waiting for their completion events."""
        if not self._chain_enabled or self.stopping is not None:
            return
        while self._chain_candidates:
            cands = self._chain_candidates
            self._chain_candidates = set()
            self._try_chain_over(cands)

    def _try_chain_over(self, cands: Set[str]) -> None:
        for tid in sorted(cands):
            call = self.calls.get(tid)
            if call is None or tid in self.dispatched:
                continue
            if call.env.provisioning.effective_gpu_count > 1:
                continue  # gangs stay on the completion-driven path
            cid = self.stream_components.get(tid)
            if cid is not None and not self.components[cid]["broken"]:
                continue  # folded into a streamed plan, never chained
            deps = self.task_deps.get(tid, [])
            if not deps or any(d not in self.dispatched for d in deps):
                continue
            ranks = set()
            inflight_deps = []
            ok = True
            for d in deps:
                dr = self.task_ranks.get(d)
                if dr is None or len(dr) != 1 or d in (
                    self.gang_pending
                ):
                    ok = False
                    break
                ranks.add(dr[0])
                if d not in self.dag_completed and d not in self.done_pending:
                    inflight_deps.append(d)
            if not ok or len(ranks) != 1 or not inflight_deps:
                continue  # multi-rank deps (or all complete): normal path
            r = next(iter(ranks))
            if r not in self.outstanding:
                continue  # rank excluded (death)
            pending_local = {
                eid
                for eid in call.input_entry_ids()
                if self.entry_producer.get(eid) in inflight_deps
            }
            self._dispatch(tid, chain_rank=r, pending_local=pending_local)
            METRICS.inc("lzy_chain_dispatches")

    def _release_gang(self, task_id: str) -> None:
        """A gang finished: its ranks are free — re-try every deferred
        gang (more than one may now fit disjoint slots)."""
        if self.deferred_gangs:
            pending = self.deferred_gangs
            self.deferred_gangs = []  # _dispatch re-defers what still can't fit
            for tid in pending:
                self._dispatch(tid)

    def _record_outputs(self, call: "LzyCall", rank: int, result: TaskResult) -> None:
        snap = self.workflow.snapshot
        for wire in result.outputs:
            meta = EntryMeta.from_wire(wire)
            meta.owners = {rank}
            inline = wire.get("inline")
            if inline is not None and rank != 0:
                # small result rode the completion event: land it on the
                # driver so client materialization is a store hit, not a
                # fetch round trip
                self.pool.agent.store.put(
                    meta.entry_id, unpickle_value(inline), pickled=inline
                )
                meta.owners.add(0)
            self.meta[meta.entry_id] = meta
            if "uri" in wire:
                snap.update_entry_uri(meta.entry_id, wire["uri"])

    def _on_failed(self, rank: int, result: TaskResult) -> None:
        self.outstanding[rank] -= 1
        self._tail_logs(rank, result)
        call = self.calls[result.task_id]
        gang = self.gang_pending.get(result.task_id)
        payload = unpickle_value(result.exc_bytes)
        exc_name, exc_msg, tb = payload[0], payload[1], payload[2]
        exc_blob = payload[3] if len(payload) > 3 else None
        self.workflow.snapshot.put(call.exception_id, (exc_name, exc_msg, tb))
        err = LzyExecutionError(
            f"Op {call.callable_name} failed on rank {rank}: "
            f"{exc_name}: {exc_msg}",
            task_id=result.task_id,
            remote_traceback=tb,
        )
        if exc_blob is not None:
            try:
                # re-raise chain carries the user's exception object,
                # rebuilt without calling a possibly-incompatible __init__
                cls, args, state = unpickle_value(exc_blob)
                cause = cls.__new__(cls)
                cause.args = args
                if state:
                    cause.__dict__.update(state)
                err.__cause__ = cause
            except Exception:  # noqa: BLE001 - string triple remains
                pass
        self.errors.append(err)
        self.journal.record(result.task_id, "failed", exc_msg)
        if gang is not None:
            gang.discard(rank)
            if gang:
                return  # remaining gang members still must report
            self.gang_pending.pop(result.task_id, None)
            self._release_gang(result.task_id)
        self.inflight -= 1
        # terminal: dispatch_ts doubles as the liveness marker that the
        # run loop's stale-event guard checks
        self.task_dispatch_ts.pop(result.task_id, None)
        self.chained_waits.pop(result.task_id, None)
