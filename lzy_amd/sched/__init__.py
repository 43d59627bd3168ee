"""DAG scheduling core.

``_core`` is the C++ extension (pybind11): DAG store + ready-queue
scheduler + crash-resume journal + xxhash64.  ``Dag`` resolves to the C++
implementation when built, else to the pure-python fallback with the same
interface (the fallback keeps CPU-only test environments working; on a GPU
box the native core is expected — lzy_amd.ops enforces native presence for
the data plane, the scheduler works either way).
"""
from __future__ import annotations

try:
    from lzy_amd.sched import _core  # type: ignore[attr-defined]

    Dag = _core.Dag
    Journal = _core.Journal
    xxhash64 = _core.xxhash64
    NATIVE = True
except ImportError:  # pragma: no cover - exercised only pre-build
    from lzy_amd.sched.pydag import PyDag as Dag  # type: ignore[assignment]
    from lzy_amd.sched.pydag import PyJournal as Journal  # type: ignore[assignment]

    def xxhash64(data: bytes) -> int:
        import hashlib

        return int.from_bytes(hashlib.blake2b(data, digest_size=8).digest(), "little")

    NATIVE = False

__all__ = ["Dag", "Journal", "xxhash64", "NATIVE"]
