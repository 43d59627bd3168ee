"""Chunked/offset-resumable transport scenario — world_size 2 on gloo
(reference: slots' offset-resumable chunked Read streams,
slots-api.proto:33-46).  Run under torch.distributed.run by
tests/test_pool_distributed.py; prints CHUNKED-TRANSPORT-OK on rank 0.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from lzy_amd.channels.transport import EntryMeta, Transport, describe_value
from lzy_amd.config import Config


def main() -> None:
    dist.init_process_group("gloo")
    rank = dist.get_rank()
    Config.reset(channel_chunk_mb=1)  # 1 MiB chunks -> multi-chunk transfers
    tr = Transport(None, None, world=dist.get_world_size())

    # --- multi-chunk tensor: 3.5 MiB f32 -> 4 chunks --------------------
    n = (3 << 18) + (1 << 17)
    if rank == 0:
        t = torch.arange(n, dtype=torch.float32)
        works, keep = tr.isend_value(t, None, dst=1)
        assert len(works) == 4, f"expected 4 chunk sends, got {len(works)}"
        for w in works:
            w.wait()
    else:
        meta = describe_value("e1", torch.empty(n, dtype=torch.float32))
        works, fin = tr.irecv_value(meta, src=0)
        assert len(works) == 4, f"expected 4 chunk recvs, got {len(works)}"
        for w in works:
            w.wait()
        got = fin()
        assert torch.equal(got, torch.arange(n, dtype=torch.float32))
        assert Transport.completed_chunks(works) == 4

    dist.barrier()

    # --- offset resume: chunk 0 already delivered -----------------------
    per = tr._chunk_elems(4)
    if rank == 0:
        t = torch.arange(n, dtype=torch.float32) * 2
        works, keep = tr.isend_value(t, None, dst=1, offset_chunks=1)
        assert len(works) == 3
        for w in works:
            w.wait()
    else:
        want = torch.arange(n, dtype=torch.float32) * 2
        buf = torch.zeros(n, dtype=torch.float32)
        buf[:per] = want[:per]  # chunk 0 survived the failed attempt
        meta = describe_value("e2", buf)
        works, fin = tr.irecv_value(meta, src=0, offset_chunks=1, into=buf)
        assert len(works) == 3
        for w in works:
            w.wait()
        assert torch.equal(fin(), want)

    dist.barrier()

    # --- multi-chunk pickled object -------------------------------------
    payload = {"blob": b"x" * (2 << 20) + b"tail", "k": 7}
    if rank == 0:
        works, keep = tr.isend_value(payload, None, dst=1)
        assert len(works) >= 2
        for w in works:
            w.wait()
    else:
        from lzy_amd.channels.transport import pickle_value

        meta = describe_value("e3", payload)
        meta.nbytes = len(pickle_value(payload))
        works, fin = tr.irecv_value(meta, src=0)
        got = [w.wait() for w in works]
        val = fin()
        assert val["k"] == 7 and val["blob"][-4:] == b"tail"

    dist.barrier()

    # --- cast-on-the-wire: f32 payload travels as fp16 -------------------
    Config.reset(channel_chunk_mb=1, channel_wire_cast="fp16")
    trc = Transport(None, None, world=dist.get_world_size())
    n2 = (1 << 20) + 77  # fp16 wire ~2 MiB -> multiple 1 MiB chunks + tail
    payload_t = (torch.arange(n2, dtype=torch.float32) % 997) / 997.0
    if rank == 0:
        works, keep = trc.isend_value(payload_t, None, dst=1)
        assert keep.dtype == torch.float16, keep.dtype  # wire form
        assert len(works) >= 3, len(works)  # chunked on the wire
        for w in works:
            w.wait()
    else:
        meta = describe_value("e4", torch.empty(n2, dtype=torch.float32))
        works, fin = trc.irecv_value(meta, src=0)
        for w in works:
            w.wait()
        got = fin()
        assert got.dtype == torch.float32
        err = (got - payload_t).abs().max().item()
        assert err < 1e-3, f"wire-cast error {err}"

    dist.barrier()

    # --- chunk-pipelined streaming pair-reduce ---------------------------
    from lzy_amd.channels.streaming import streamed_reduce_pair

    n3 = (5 << 18) + 333  # ~5 MiB f32, odd tail
    mine = torch.arange(n3, dtype=torch.float32) + (1000.0 if rank else 0.0)
    if rank == 0:
        red = streamed_reduce_pair(mine, peer=1, is_receiver=True,
                                   chunk_bytes=1 << 20)
        want = (torch.arange(n3, dtype=torch.float32)
                + (torch.arange(n3, dtype=torch.float32) + 1000.0)) * 0.5
        assert torch.allclose(red, want), "streamed reduce mismatch"
    else:
        out = streamed_reduce_pair(mine, peer=0, is_receiver=False,
                                   chunk_bytes=1 << 20)
        assert out is None

    dist.barrier()
    if rank == 0:
        print("CHUNKED-TRANSPORT-OK", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
