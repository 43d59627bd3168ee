"""Per-op HIP stream placement + stream-race guard.

Two jobs:

1. **Overlap** (BASELINE north star: "overlaps independent ops on separate
   HIP streams"): each executor THREAD is bound to one of a small pool of
   HIP streams per device (round-robin at first use), so concurrently
   executing ops on the *same* GPU overlap their kernels/DMA instead of
   serializing on the legacy default stream — see next_stream() for why
   per-thread beats per-op rotation.

2. **Race detection** (SURVEY §5.2 — the reference is JVM and needs none;
   a HIP runtime does): cross-stream value hand-off is only safe through
   an event.  The placer records a ``hipEvent`` per produced entry and
   inserts ``wait_event`` on the consumer's stream at every store read.
   With ``LZY_STREAM_CHECK=1`` it additionally *verifies* the discipline:
   a device tensor read whose producing event was never recorded counts
   as a detected race (``lzy_stream_races`` metric) and raises.

CPU-only processes: every method is a no-op (streams disabled), so the
same code path runs unchanged in CPU tests.
"""
from __future__ import annotations

import os
import threading
from typing import Any, Dict, Optional, Tuple

import torch

from lzy_amd.utils.metrics import METRICS


def _check_mode() -> bool:
    return os.environ.get("LZY_STREAM_CHECK", "") not in ("", "0")


class StreamPlacer:
    """Round-robin HIP stream pool + entry-id -> completion-event registry."""

    def __init__(self) -> None:
        from collections import deque

        self._lock = threading.Lock()
        self._streams: Dict[int, list] = {}  # device index -> [Stream]
        self._rr: Dict[int, int] = {}
        self._events: Dict[str, Tuple[Any, Any]] = {}  # entry -> (event, stream)
        # entries whose event was deliberately retired after a host-side
        # completion guarantee (workflow exit / store drop): a later read
        # is safe and must not count as a race in strict mode
        self._retired: set = set()
        self._retired_order: deque = deque()
        self._tls = threading.local()
        self.enabled = torch.cuda.is_available()

    def _pool(self, device: int) -> list:
        with self._lock:
            if device not in self._streams:
                from lzy_amd.config import get_config

                n = getattr(get_config(), "op_streams", 4) or 4
                self._streams[device] = [
                    torch.cuda.Stream(device=device) for _ in range(n)
                ]
                self._rr[device] = 0
            return self._streams[device]

    def next_stream(self) -> Optional[torch.cuda.Stream]:
        """Stream for ops on the *current thread* (None on CPU).

        Streams are bound per executor thread, round-robin at first use —
        NOT rotated per op: sequential ops in one thread share a stream
        (torch's caching allocator is per-stream; rotation fragments it
        and un-orders hidden op state like persistent model/optimizer
        tensors), while genuinely concurrent executor threads land on
        distinct streams and overlap."""
        if not self.enabled:
            return None
        dev = torch.cuda.current_device()
        by_dev = getattr(self._tls, "stream_by_dev", None)
        if by_dev is None:
            by_dev = self._tls.stream_by_dev = {}
        s = by_dev.get(dev)
        if s is None:
            pool = self._pool(dev)
            with self._lock:
                i = self._rr[dev]
                self._rr[dev] = (i + 1) % len(pool)
            s = by_dev[dev] = pool[i]
        return s

    # -- producer side ------------------------------------------------------

    def record_output(self, entry_id: str, value: Any,
                      stream: Optional[torch.cuda.Stream] = None,
                      if_absent: bool = False) -> None:
        """Record a completion event for a device-tensor entry produced on
        ``stream`` (default: the current stream of the value's device).

        ``if_absent``: keep an existing (more precise, producer-stream)
        event — used by the blanket store-put hook, which runs on the
        publishing thread's current stream."""
        if not self.enabled or not isinstance(value, torch.Tensor) or not value.is_cuda:
            return
        if if_absent:
            with self._lock:
                if entry_id in self._events:
                    return
        s = stream if stream is not None else torch.cuda.current_stream(value.device)
        ev = torch.cuda.Event()
        ev.record(s)
        with self._lock:
            if if_absent and entry_id in self._events:
                return
            self._events[entry_id] = (ev, s)
        METRICS.inc("lzy_stream_events_recorded")

    # -- consumer side ------------------------------------------------------

    def wait_value(self, entry_id: str, value: Any) -> None:
        """Make the current stream wait for the entry's producing event.
        Device-side wait: does not block the host."""
        if not self.enabled or not isinstance(value, torch.Tensor) or not value.is_cuda:
            return
        with self._lock:
            hit = self._events.get(entry_id)
            retired = entry_id in self._retired
        if hit is None:
            if _check_mode() and not retired:
                METRICS.inc("lzy_stream_races")
                raise RuntimeError(
                    f"stream-race check: device tensor entry {entry_id} read "
                    f"with no recorded producing event"
                )
            return
        ev, src_stream = hit
        cur = torch.cuda.current_stream(value.device)
        if cur != src_stream:
            cur.wait_event(ev)
            METRICS.inc("lzy_stream_waits_inserted")

    def _retire(self, entry_id: str) -> None:
        # assumes self._lock held
        if entry_id not in self._retired:
            self._retired.add(entry_id)
            self._retired_order.append(entry_id)
            while len(self._retired_order) > 100_000:
                self._retired.discard(self._retired_order.popleft())

    def drop(self, entry_id: str) -> None:
        with self._lock:
            if self._events.pop(entry_id, None) is not None:
                self._retire(entry_id)

    def sync_and_drop(self, entry_id: str) -> None:
        """Host-side completion guarantee + registry cleanup (workflow
        exit: values may outlive the workflow as plain tensors)."""
        with self._lock:
            hit = self._events.pop(entry_id, None)
            self._retire(entry_id)
        if hit is not None:
            hit[0].synchronize()

    def clear(self) -> None:
        with self._lock:
            self._events.clear()

    def stats(self) -> Dict[str, float]:
        return {
            "recorded": METRICS.counter_value("lzy_stream_events_recorded"),
            "waits": METRICS.counter_value("lzy_stream_waits_inserted"),
            "races": METRICS.counter_value("lzy_stream_races"),
        }


STREAMS = StreamPlacer()
