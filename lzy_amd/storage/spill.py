"""Pinned-host spill tier for the HBM-resident result store.

BASELINE north star: "whiteboard/s3-sink become an HBM-resident result
store sized for 288 GB per GPU with pinned-host async spill".  Values
live in HBM as long as they fit; when allocated HBM crosses the
threshold, least-recently-used store entries are copied to *pinned* host
buffers on a dedicated HIP stream (async DMA — compute streams keep
running) and the device copy is released.  First access of a spilled
entry brings it back with one H2D DMA.

The reference's equivalent tier is S3 multipart upload through the
vendored transmitter (util-s3 `ru.yandex.qe.s3.transfer`); here the
"network" is the host DMA engine at ~50 GB/s instead of a cloud object
store at ~100 MB/s.

CPU-testable: the LRU/threshold bookkeeping takes injectable
``bytes_in_use``/``capacity`` callables and a ``mover``; the real GPU
path is exercised by tests/test_gpu_workflow.py.
"""
from __future__ import annotations

import logging
import threading
from collections import OrderedDict
from typing import Any, Callable, Dict, Optional, Tuple

import torch

from lzy_amd.utils.metrics import METRICS

_LOG = logging.getLogger("lzy_amd.spill")


class _DefaultMover:
    """Real device<->pinned-host mover on a dedicated stream."""

    def __init__(self, device: Optional[torch.device]):
        self.device = device
        self._stream = (
            torch.cuda.Stream(device=device)
            if device is not None and torch.cuda.is_available()
            else None
        )

    def to_host(self, t: torch.Tensor, entry_id: str = "") -> Tuple[torch.Tensor, Any]:
        """Async D2H into pinned memory; returns (host_tensor, ready_event)."""
        if self._stream is None:
            return t.cpu(), None
        from lzy_amd.runtime.streams import STREAMS

        with torch.cuda.stream(self._stream):
            if entry_id:
                STREAMS.wait_value(entry_id, t)  # order D2H after producer
            host = torch.empty_like(t, device="cpu", pin_memory=True)
            host.copy_(t, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record(self._stream)
        return host, ev

    def to_device(self, host: torch.Tensor, device: torch.device) -> torch.Tensor:
        if self._stream is None:
            return host.to(device) if device is not None else host
        with torch.cuda.stream(self._stream):
            t = host.to(device, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record(self._stream)
        # consumer ordering: current stream waits for the H2D
        torch.cuda.current_stream(device).wait_event(ev)
        return t


class SpillManager:
    """LRU spill of device-tensor store entries past an HBM threshold."""

    def __init__(
        self,
        device: Optional[torch.device] = None,
        threshold_frac: Optional[float] = None,
        bytes_in_use: Optional[Callable[[], int]] = None,
        capacity: Optional[Callable[[], int]] = None,
        mover=None,
    ):
        from lzy_amd.config import get_config

        cfg = get_config()
        self.device = device
        self.threshold_frac = (
            threshold_frac
            if threshold_frac is not None
            else getattr(cfg, "spill_threshold_frac", 0.85)
        )
        self.enabled = getattr(cfg, "spill_enabled", True)
        if bytes_in_use is None:
            if torch.cuda.is_available() and device is not None:
                bytes_in_use = lambda: torch.cuda.memory_allocated(device)  # noqa: E731
                capacity = lambda: torch.cuda.get_device_properties(  # noqa: E731
                    device
                ).total_memory
            else:
                bytes_in_use = lambda: 0  # noqa: E731
                capacity = lambda: 1  # noqa: E731
        self._bytes_in_use = bytes_in_use
        self._capacity = capacity
        self._mover = mover if mover is not None else _DefaultMover(device)
        self._lock = threading.Lock()
        self._lru: "OrderedDict[str, int]" = OrderedDict()  # entry -> nbytes
        self._spilled: Dict[str, Tuple[torch.Tensor, torch.device, Any]] = {}
        self._capacity_cache: Optional[int] = None
        # allowance for HBM the store does not track (model weights,
        # activations): the real allocator is consulted once tracked
        # bytes approach the threshold
        self._untracked_slack = 32 << 30

    # -- bookkeeping --------------------------------------------------------

    def track(self, entry_id: str, value: Any) -> None:
        """Register/touch a store entry (call at put and get)."""
        if not getattr(value, "is_cuda", False):  # duck-typed for tests
            return
        with self._lock:
            self._lru.pop(entry_id, None)
            self._lru[entry_id] = value.numel() * value.element_size()

    def forget(self, entry_id: str) -> None:
        with self._lock:
            self._lru.pop(entry_id, None)
            self._spilled.pop(entry_id, None)

    def is_spilled(self, entry_id: str) -> bool:
        with self._lock:
            return entry_id in self._spilled

    @property
    def spilled_count(self) -> int:
        with self._lock:
            return len(self._spilled)

    # -- spill / unspill ----------------------------------------------------

    def maybe_spill(self, store_values: Dict[str, Any]) -> int:
        """Spill LRU entries while HBM use exceeds the threshold.
        Returns the number of entries spilled."""
        if not self.enabled:
            return 0
        # fast path: querying the real allocator (torch memory_stats walks
        # the whole stats tree, ~0.1 ms) is pointless while the store's own
        # tracked bytes are far below the threshold
        if self._capacity_cache is None:
            self._capacity_cache = self._capacity()
        limit = int(self._capacity_cache * self.threshold_frac)
        with self._lock:
            tracked = sum(self._lru.values())
        if tracked < limit // 2 and tracked + self._untracked_slack < limit:
            return 0
        n = 0
        while self._bytes_in_use() > limit:
            with self._lock:
                victim = next(iter(self._lru), None)
                if victim is not None:
                    self._lru.pop(victim, None)
            if victim is None:
                break
            value = store_values.get(victim)
            if not getattr(value, "is_cuda", False):
                continue
            try:
                host, ev = self._mover.to_host(value, victim)
            except TypeError:  # custom test movers may take one arg
                host, ev = self._mover.to_host(value)
            if ev is not None:
                ev.synchronize()  # device copy may be freed only after D2H
            with self._lock:
                # the _spilled insert and the store swap must be ONE
                # atomic step: a concurrent get()/unspill between them
                # would unspill (no-op on store) and then be overwritten
                # by the host tensor below, stranding a CPU tensor in a
                # device entry.  Also abort if the entry was dropped or
                # replaced while the D2H was in flight.
                if store_values.get(victim) is not value:
                    continue
                self._spilled[victim] = (host, value.device, ev)
                store_values[victim] = host  # device ref dropped -> freed
            nbytes = value.numel() * value.element_size()
            METRICS.inc("lzy_spill_out")
            METRICS.inc("lzy_spill_bytes", nbytes)
            n += 1
            _LOG.info("spilled entry %s (%d bytes) to pinned host", victim, nbytes)
        return n

    def unspill(self, entry_id: str, store_values: Dict[str, Any]) -> Optional[torch.Tensor]:
        """Bring a spilled entry back to its device; returns the device
        tensor (and re-registers it in the LRU).  Returns None when the
        entry is not spilled — including when a concurrent unspill won the
        race, in which case ``store_values`` already holds the device
        tensor by the time the lock is released here.

        Fully locked (pop + H2D + store write): the H2D is a rare slow
        path, and partial visibility (popped from _spilled but not yet
        back in the store) would let maybe_spill/get interleave wrongly.
        """
        with self._lock:
            hit = self._spilled.pop(entry_id, None)
            if hit is None:
                return None
            host, device, _ = hit
            t = self._mover.to_device(host, device)
            # re-publish a producing event BEFORE publication (the H2D is
            # ordered into the current stream by the mover)
            from lzy_amd.runtime.streams import STREAMS

            if isinstance(t, torch.Tensor) and t.is_cuda:
                STREAMS.record_output(entry_id, t)
            store_values[entry_id] = t
            # inline LRU touch (track() re-acquires this lock)
            if getattr(t, "is_cuda", False):
                self._lru.pop(entry_id, None)
                self._lru[entry_id] = t.numel() * t.element_size()
        METRICS.inc("lzy_spill_in")
        return t
