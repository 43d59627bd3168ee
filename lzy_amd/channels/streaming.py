"""Chunk-pipelined streaming reduction over a p2p channel.

The pairwise tree-merge of a large shard costs one full transfer per
level plus the combine.  Streaming overlaps them: the receiver posts the
recv for chunk k+1 while combining chunk k (double buffer), so the
combine rides inside the transfer's shadow and the pair-reduce finishes
in ~transfer time.  On device the combine is the fused axpby kernel; on
host it's a torch op.

Round-2 note: the scheduler-integrated version of this idea is
``channels/treeplan.py`` (whole merge TREES fold into one chunk-major
plan with cross-level forwarding).  This standalone pair primitive
remains the minimal testable form (chunked_transport_script) and the
reference point the plan generalizes.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist


def streamed_reduce_pair(
    t: torch.Tensor,
    peer: int,
    is_receiver: bool,
    alpha: float = 0.5,
    beta: float = 0.5,
    chunk_bytes: int = 64 << 20,
    group: Optional[dist.ProcessGroup] = None,
) -> Optional[torch.Tensor]:
    """Pairwise reduce of ``t`` with ``peer``'s tensor.

    Receiver returns ``alpha*t + beta*peer_t`` (new tensor); sender sends
    its ``t`` in chunks and returns None.  Both sides must pass the same
    shape/dtype/chunk_bytes.
    """
    flat = t.detach().contiguous().view(-1)
    n = flat.numel()
    per = max(1, chunk_bytes // max(1, flat.element_size()))
    starts = list(range(0, n, per))

    if not is_receiver:
        works = [
            dist.isend(flat[s: min(s + per, n)], dst=peer, group=group)
            for s in starts
        ]
        for w in works:
            w.wait()
        return None

    out = torch.empty_like(flat)
    bufs = [
        torch.empty(min(per, n), dtype=flat.dtype, device=flat.device)
        for _ in range(2)
    ]

    def post(i: int):
        s = starts[i]
        ln = min(s + per, n) - s
        return dist.irecv(bufs[i % 2][:ln], src=peer, group=group), s, ln

    use_hip = flat.is_cuda
    if use_hip:
        from lzy_amd import ops as _ops

        use_hip = _ops.NATIVE and flat.dtype in (
            torch.float32, torch.float16, torch.bfloat16
        )

    inflight = post(0)
    for i in range(len(starts)):
        w, s, ln = inflight
        nxt = post(i + 1) if i + 1 < len(starts) else None
        w.wait()
        src_c = flat[s: s + ln]
        dst_c = out[s: s + ln]
        if use_hip:
            from lzy_amd import ops as _ops

            _ops.axpby(src_c, bufs[i % 2][:ln], alpha, beta, dst=dst_c)
        else:
            torch.add(src_c * alpha, bufs[i % 2][:ln], alpha=beta, out=dst_c)
        inflight = nxt
    return out.view(t.shape)
