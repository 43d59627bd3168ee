"""Chunk-streamed merge-tree plans.

The DAG's pairwise merge tree (ops declared ``@op(pair_reduce=(α, β))``,
computing ``α·a + β·b`` for two same-shape tensors) is the transfer-bound
part of a fan-out/reduce workflow: executed op-by-op, every tree level
costs one full xGMI transfer before its combine can start, so an N-leaf
tree pays ~log2(N) transfer latencies end to end.

A *streamed tree plan* collapses that to ~one transfer: the driver folds
a connected component of pair-reduce tasks into a single multi-rank plan
executed as a custom point-to-point collective — every rank walks the
chunk-major schedule ``for chunk j: for node (topo order): recv j →
combine j → forward j``, so a chunk flows through ALL tree levels
back-to-back while later chunks are still in flight on the first level.
Total latency ≈ one shard transfer + (levels-1) chunk times.  Interior
node outputs still materialize on their compute rank (a per-level
accumulator buffer IS the value), so every op result stays readable —
no semantic narrowing vs op-by-op execution.

Reference analogue: chunked offset-resumable slot streams
(slots-api.proto:33-46) and the SURVEY §5.8 note that ring collectives
are per-link bound on xGMI — this plan schedules the tree's level-0
pairs onto disjoint xGMI links in parallel and pipelines across levels.

Ordering discipline: all plan p2p runs on a DEDICATED process group
(pg_stream) so it can be issued from executor threads without
interleaving with the serve-loop-issued transfer traffic on pg_data;
the driver keeps at most one plan active pool-wide, and per (src, dst)
pair every rank issues chunk-major in global topo order — pairwise
consistent by construction (checked at build time).
"""
from __future__ import annotations

import logging
import math
import time
from typing import Any, Dict, List, Optional, Sequence, Set, Tuple

import torch
import torch.distributed as dist

from lzy_amd.utils.metrics import METRICS

_LOG = logging.getLogger("lzy_amd.treeplan")


# ---------------------------------------------------------------------------
# driver side: component discovery + plan construction
# ---------------------------------------------------------------------------

def _gpu_count(call: Any) -> int:
    try:
        return int(call.env.provisioning.effective_gpu_count)
    except AttributeError:
        return 0


def find_components(calls: Dict[str, Any]) -> List[List[str]]:
    """Connected components of pair_reduce tasks (edges: one task's output
    consumed by another), each topo-ordered.  Only components with >= 2
    members are worth folding (a single pair is one transfer either way).

    ``calls``: task_id -> LzyCall (must expose .pair_reduce, .entry_ids,
    .input_entry_ids(), .kwarg_entry_ids).
    """
    members = {
        tid for tid, c in calls.items()
        if getattr(c, "pair_reduce", None) is not None
        and len(c.entry_ids) == 1
        and len(c.input_entry_ids()) == 2
        and not c.kwarg_entry_ids
        # cached ops keep the op-by-op path: the plan executor bypasses
        # the result cache (both lookup and write), and silently ignoring
        # @op(cache=True) would break the caching contract
        and not getattr(c, "cache", False)
        # gang-provisioned ops keep gang semantics (op body on k ranks)
        and _gpu_count(c) <= 1
    }
    if not members:
        return []
    producer: Dict[str, str] = {}
    for tid in members:
        for eid in calls[tid].entry_ids:
            producer[eid] = tid
    # adjacency within the member set
    adj: Dict[str, Set[str]] = {tid: set() for tid in members}
    deps: Dict[str, Set[str]] = {tid: set() for tid in members}
    for tid in members:
        for eid in calls[tid].input_entry_ids():
            src = producer.get(eid)
            if src is not None and src in members:
                adj[src].add(tid)
                adj[tid].add(src)
                deps[tid].add(src)
    out: List[List[str]] = []
    seen: Set[str] = set()
    for start in sorted(members):
        if start in seen:
            continue
        comp: Set[str] = set()
        stack = [start]
        while stack:
            t = stack.pop()
            if t in comp:
                continue
            comp.add(t)
            stack.extend(adj[t] - comp)
        seen |= comp
        if len(comp) < 2:
            continue
        # topo order within the component (deps are intra-component)
        order: List[str] = []
        ready = sorted(t for t in comp if not (deps[t] & comp))
        indeg = {t: len(deps[t] & comp) for t in comp}
        while ready:
            t = ready.pop(0)
            order.append(t)
            for nxt in sorted(adj[t]):
                if t in deps.get(nxt, ()):
                    indeg[nxt] -= 1
                    if indeg[nxt] == 0:
                        ready.append(nxt)
        if len(order) == len(comp):
            out.append(order)
    return out


def build_plan(
    plan_id: str,
    member_order: Sequence[str],
    calls: Dict[str, Any],
    meta_of,
    chunk_bytes: int,
    cuda_p2p: bool,
) -> Optional[dict]:
    """Construct the per-rank step schedule for one component.

    ``meta_of(eid)`` -> EntryMeta for externally-produced entries (must
    have confirmed owners); returns None if the component cannot be
    safely folded (shape/dtype mismatch, missing owner, or per-pair
    order inconsistency) — the caller then falls back to op-by-op
    execution.
    """
    out_entry_of: Dict[str, str] = {
        tid: calls[tid].entry_ids[0] for tid in member_order
    }
    internal_out: Dict[str, str] = {e: t for t, e in out_entry_of.items()}

    # reference shape/dtype from the first external leaf
    shape: Optional[Tuple[int, ...]] = None
    dtype: Optional[str] = None
    device_cuda = False
    node_rank: Dict[str, int] = {}
    # step records per rank, tagged with global topo index for ordering
    steps_by_rank: Dict[int, List[dict]] = {}
    # per ordered pair: (producer_issue_topo, consumer_topo) — used to
    # verify sender issue order matches receiver post order
    pair_edges: Dict[Tuple[int, int], List[Tuple[int, int]]] = {}

    def add_step(rank: int, topo: int, step: dict) -> None:
        step["topo"] = topo
        steps_by_rank.setdefault(rank, []).append(step)

    topo_idx = 0
    for tid in member_order:
        call = calls[tid]
        a_eid, b_eid = call.input_entry_ids()
        alpha, beta = call.pair_reduce
        srcs = []
        for eid in (a_eid, b_eid):
            if eid in internal_out:
                child = internal_out[eid]
                srcs.append(("node", eid, node_rank[child]))
            else:
                m = meta_of(eid)
                if m is None or m.kind != "tensor" or not m.owners:
                    return None
                if shape is None:
                    shape, dtype = tuple(m.shape), m.dtype
                    device_cuda = m.device_type == "cuda"
                elif tuple(m.shape) != shape or m.dtype != dtype:
                    return None
                if (m.device_type == "cuda") != device_cuda:
                    return None
                owner = 0 if 0 in m.owners else sorted(m.owners)[0]
                srcs.append(("entry", eid, owner))
        if shape is None:
            return None  # all-internal first node cannot happen (topo order)
        # place on the first internal child's rank, else first owner
        internal = [s for s in srcs if s[0] == "node"]
        my_rank = internal[0][2] if internal else srcs[0][2]
        node_rank[tid] = my_rank
        local_srcs = [s for s in srcs if s[2] == my_rank]
        remote_srcs = [s for s in srcs if s[2] != my_rank]
        if len(remote_srcs) > 1:
            return None  # placement guarantees <= 1, keep the invariant
        step = {
            "op": "node",
            "task": tid,
            "out": out_entry_of[tid],
            "alpha": float(alpha),
            "beta": float(beta),
            "local": local_srcs[0][1],
            "local_is_entry": local_srcs[0][0] == "entry",
            "remote_src": None,
            "local2": None,
            "local2_is_entry": False,
            "send_to": [],
        }
        if remote_srcs:
            kind, eid, src_rank = remote_srcs[0]
            step["remote_src"] = src_rank
            if kind == "entry":
                # leaf owner streams the entry's chunks to us; the send
                # step carries the consumer's topo index, so its issue
                # position equals the edge position
                add_step(src_rank, topo_idx,
                         {"op": "leaf_send", "entry": eid, "dst": my_rank})
                issue_topo = topo_idx
            else:
                # producing node forwards its accumulator chunks as they
                # are combined — issued at the CHILD's topo position
                child = internal_out[eid]
                issue_topo = member_order.index(child)
                for st in steps_by_rank[src_rank]:
                    if st.get("task") == child:
                        # (destination rank, consumer topo): the tag the
                        # receiver posts is derived from ITS node's topo
                        st["send_to"].append([my_rank, topo_idx])
                        break
            pair_edges.setdefault((src_rank, my_rank), []).append(
                (issue_topo, topo_idx)
            )
        else:
            step["local2"] = local_srcs[1][1]
            step["local2_is_entry"] = local_srcs[1][0] == "entry"
        add_step(my_rank, topo_idx, step)
        topo_idx += 1

    # per-pair order consistency: RCCL/gloo match p2p by ISSUE ORDER per
    # (src, dst) pair — the sender walks its steps in its topo order and
    # the receiver posts in its own topo order, so both orders over the
    # shared edges must agree.  For trees they always do (a child's topo
    # precedes its consumer's); crossing orders are possible in exotic
    # DAGs — refuse to fold those (caller falls back to op-by-op).
    for pair, es in pair_edges.items():
        by_producer = sorted(es)
        by_consumer = sorted(es, key=lambda pc: pc[1])
        if by_producer != by_consumer:
            _LOG.info("stream plan %s: crossing edge order on pair %s — "
                      "falling back", plan_id, pair)
            return None
    for rank in steps_by_rank:
        steps_by_rank[rank].sort(key=lambda s: s["topo"])

    numel = 1
    for s in shape:
        numel *= s
    elem = torch.empty(0, dtype=getattr(torch, dtype)).element_size()
    per = max(1, chunk_bytes // elem)
    # pipelining needs several chunks; cap chunk size so a shard splits
    # into at least 4 (but never below 1 MiB of elements)
    if numel // per < 4:
        per = max((1 << 20) // elem, math.ceil(numel / 4))
    return {
        "plan_id": plan_id,
        # per-plan tag base: edge tag = (nonce + topo) so a FAILED
        # plan's stale posted recvs on pg_stream can never match a later
        # plan's chunks (gloo matches tags; RCCL ignores them)
        "nonce": abs(hash(plan_id)) % (1 << 20) << 9,
        "shape": list(shape),
        "dtype": dtype,
        "numel": numel,
        "chunk_elems": per,
        "device_cuda": device_cuda,
        "cuda_p2p": bool(cuda_p2p),
        "steps_by_rank": steps_by_rank,
        "participants": sorted(steps_by_rank),
        "node_rank": node_rank,
        "member_order": list(member_order),
    }


# ---------------------------------------------------------------------------
# worker side: plan execution
# ---------------------------------------------------------------------------

def run_stream_plan(
    plan: dict,
    my_steps: List[dict],
    store,
    pg,
    device,
    wait_timeout: float = 120.0,
) -> List[dict]:
    """Execute one rank's schedule; returns wire metas of the node
    outputs this rank computed (already landed in the store).

    Chunk-major walk: ``for j: for step (topo order)``.  Recvs are posted
    lazily with a lookahead window in per-src order (matching the
    sender's issue order — RCCL/gloo pair messages by order, not tags);
    sends are non-blocking and awaited at the end.
    """
    from lzy_amd.channels.transport import describe_value
    from lzy_amd.runtime.streams import STREAMS

    shape = tuple(plan["shape"])
    dtype = getattr(torch, plan["dtype"])
    numel, per = plan["numel"], plan["chunk_elems"]
    nchunks = max(1, math.ceil(numel / per))
    cuda_p2p = plan["cuda_p2p"]
    want_cuda = plan["device_cuda"] and device is not None
    on_device = want_cuda and cuda_p2p
    host_stage = want_cuda and not cuda_p2p  # oversubscribed harnesses

    def flat_entry(eid: str) -> torch.Tensor:
        if not store.has(eid) and not store.wait_present(eid, timeout=wait_timeout):
            raise RuntimeError(f"stream plan: entry {eid} never arrived")
        v = store.get(eid)
        if not isinstance(v, torch.Tensor):
            raise RuntimeError(f"stream plan: entry {eid} is not a tensor")
        return v.detach().contiguous().view(-1)

    outs: Dict[str, torch.Tensor] = {}     # out entry -> flat accumulator
    local_flat: Dict[int, torch.Tensor] = {}   # step topo -> local source
    local2_flat: Dict[int, torch.Tensor] = {}
    leaf_flat: Dict[int, torch.Tensor] = {}

    import datetime as _dt

    recv_timeout = _dt.timedelta(seconds=wait_timeout)

    stats = {"recv_wait_s": 0.0, "sends": 0, "recvs": 0}

    def wait_recv(w) -> None:
        # gloo honors the timeout (raises instead of hanging on a dead
        # peer); the RCCL path is stream-ordered — plain wait
        t0 = time.perf_counter()
        if on_device:
            w.wait()
        else:
            w.wait(recv_timeout)
        stats["recv_wait_s"] += time.perf_counter() - t0
        stats["recvs"] += 1

    stream = STREAMS.next_stream()
    from contextlib import nullcontext

    with (torch.cuda.stream(stream) if stream is not None else nullcontext()):
        # resolve entry-sourced tensors up front (waits for settles) and
        # allocate output accumulators
        for st in my_steps:
            if st["op"] == "leaf_send":
                leaf_flat[st["topo"]] = flat_entry(st["entry"])
                STREAMS.wait_value(st["entry"], leaf_flat[st["topo"]])
            else:
                if st["local_is_entry"]:
                    local_flat[st["topo"]] = flat_entry(st["local"])
                    STREAMS.wait_value(st["local"], local_flat[st["topo"]])
                if st["local2"] is not None and st["local2_is_entry"]:
                    local2_flat[st["topo"]] = flat_entry(st["local2"])
                    STREAMS.wait_value(st["local2"], local2_flat[st["topo"]])
                src_t = (
                    local_flat[st["topo"]]
                    if st["local_is_entry"] else outs[st["local"]]
                )
                outs[st["out"]] = torch.empty(
                    numel, dtype=dtype, device=src_t.device
                )

        # recv machinery: per-src ordered sequence of (chunk j, step),
        # posted with a lookahead window
        seq_by_src: Dict[int, List[dict]] = {}
        for st in my_steps:
            if st["op"] == "node" and st["remote_src"] is not None:
                seq_by_src.setdefault(st["remote_src"], []).append(st)
        recv_buf: Dict[int, torch.Tensor] = {}  # step topo -> full buffer
        for src, sts in seq_by_src.items():
            for st in sts:
                recv_buf[st["topo"]] = torch.empty(
                    numel, dtype=dtype, device=device if on_device else None
                )
        posted: Dict[int, int] = {s: 0 for s in seq_by_src}
        recv_works: Dict[Tuple[int, int], Any] = {}  # (topo, j) -> work
        WINDOW = 4

        def chunk_bounds(j: int) -> Tuple[int, int]:
            s = j * per
            return s, min(s + per, numel)

        nonce = plan.get("nonce", 0)

        def edge_tag(topo: int) -> int:
            # gloo-only matching aid; the RCCL path is order-matched and
            # torch documents tags as unsupported there
            if on_device:
                return 0
            return (nonce + topo) % (1 << 30)

        def post_up_to(src: int, upto_flat_idx: int) -> None:
            sts = seq_by_src[src]
            width = len(sts)
            while posted[src] <= min(upto_flat_idx + WINDOW,
                                     nchunks * width - 1):
                k = posted[src]
                j, st = divmod(k, width)
                st = sts[st]
                s, e = chunk_bounds(j)
                recv_works[(st["topo"], j)] = dist.irecv(
                    recv_buf[st["topo"]][s:e], src=src, group=pg,
                    tag=edge_tag(st["topo"]),
                )
                posted[src] += 1

        send_works: List[Any] = []
        staged: List[torch.Tensor] = []  # host staging keepalive

        def issue_send(t: torch.Tensor, dst: int, topo: int) -> None:
            stats["sends"] += 1
            tg = edge_tag(topo)
            if host_stage and t.is_cuda:
                c = t.to("cpu")
                staged.append(c)
                send_works.append(dist.isend(c, dst=dst, group=pg, tag=tg))
            else:
                send_works.append(
                    dist.isend(t.contiguous(), dst=dst, group=pg, tag=tg)
                )

        try:
            from lzy_amd import ops as _ops

            use_hip = _ops.NATIVE
        except Exception:  # pragma: no cover - ops ext missing
            use_hip = False

        def combine(dst_c, a_c, b_c, alpha, beta):
            if (
                dst_c.is_cuda and use_hip
                and dst_c.dtype in (torch.float32, torch.float16, torch.bfloat16)
            ):
                from lzy_amd import ops as _ops

                _ops.axpby(a_c, b_c, alpha, beta, dst=dst_c)
            else:
                torch.add(a_c * alpha, b_c, alpha=beta, out=dst_c)

        # ---- the chunk-major walk ---------------------------------------
        for j in range(nchunks):
            s, e = chunk_bounds(j)
            for st in my_steps:
                if st["op"] == "leaf_send":
                    issue_send(leaf_flat[st["topo"]][s:e], st["dst"],
                               st["topo"])
                    continue
                a = (
                    local_flat[st["topo"]]
                    if st["local_is_entry"] else outs[st["local"]]
                )
                out = outs[st["out"]]
                if st["remote_src"] is not None:
                    src = st["remote_src"]
                    sts = seq_by_src[src]
                    flat_idx = j * len(sts) + sts.index(st)
                    post_up_to(src, flat_idx)
                    w = recv_works.pop((st["topo"], j))
                    wait_recv(w)
                    b_c = recv_buf[st["topo"]][s:e]
                    if host_stage:
                        b_c = b_c.to(a.device)
                    combine(out[s:e], a[s:e], b_c, st["alpha"], st["beta"])
                else:
                    b = (
                        local2_flat[st["topo"]]
                        if st["local2_is_entry"] else outs[st["local2"]]
                    )
                    combine(out[s:e], a[s:e], b[s:e], st["alpha"], st["beta"])
                for dst, fwd_topo in st["send_to"]:
                    issue_send(out[s:e], dst, fwd_topo)

        for w in send_works:
            w.wait()

        # materialize this rank's node outputs (per-level accumulators ARE
        # the op results — every interior value stays readable)
        results: List[dict] = []
        for st in my_steps:
            if st["op"] != "node":
                continue
            value = outs[st["out"]].view(shape)
            STREAMS.record_output(st["out"], value, stream=stream)
            store.put(st["out"], value)
            meta = describe_value(st["out"], value)
            results.append({**meta.to_wire(), "task": st["task"]})
        METRICS.inc("lzy_stream_plan_nodes", len(results))
        # per-rank telemetry rides the first result (or a bare record):
        # the driver aggregates recv-wait across ranks for the SCALE
        # config block (overlap evidence)
        if results:
            results[0]["plan_stats"] = dict(stats)
    return results
