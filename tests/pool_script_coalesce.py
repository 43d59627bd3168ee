"""The full pool transfer battery under a SIMULATED NCCL-coalesced
batch_isend_irecv (one work per whole group, as RCCL returns on the
real multi-GPU node): Transport.issue must normalize works one-per-op
so settle slicing and chunk accounting stay correct.  This exercises,
end to end, the exact works-shape the 1-GPU harnesses can never produce
(oversubscription forces gloo, which returns per-op works).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch.distributed as dist

_orig = dist.batch_isend_irecv


class _GroupWork:
    """One work for a whole batch, NCCL-coalescing style."""

    def __init__(self, ws):
        self._ws = ws

    def wait(self, *a, **k):
        for w in self._ws:
            w.wait(*a, **k)
        return True

    def is_completed(self):
        return all(w.is_completed() for w in self._ws)


def _coalesced(ops):
    ws = _orig(ops)
    return [_GroupWork(ws)] if ws else []


dist.batch_isend_irecv = _coalesced

import tests.pool_script as base  # noqa: E402  (after the patch)

if __name__ == "__main__":
    base.main()
