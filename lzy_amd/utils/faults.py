"""Fault injection points.

Reference capability (lzy-service debug/InjectedFailures.java:9-45 and
peers in graph-executor-2 / allocator / channel-manager): indexed failure
points toggled at runtime, used by restart/crash-resume tests.  Here a
failure point is armed by name with a countdown; ``hit()`` raises (or
kills the process, to exercise journal resume) when the countdown reaches
zero.
"""
from __future__ import annotations

import os
import threading
from typing import Dict


class InjectedFailure(Exception):
    pass


class _Faults:
    def __init__(self) -> None:
        self._armed: Dict[str, int] = {}
        self._kind: Dict[str, str] = {}
        self._lock = threading.Lock()

    def arm(self, point: str, countdown: int = 0, kind: str = "raise") -> None:
        """kind: 'raise' -> InjectedFailure; 'exit' -> os._exit(42) (crash)."""
        with self._lock:
            self._armed[point] = countdown
            self._kind[point] = kind

    def disarm(self, point: str) -> None:
        with self._lock:
            self._armed.pop(point, None)
            self._kind.pop(point, None)

    def clear(self) -> None:
        with self._lock:
            self._armed.clear()
            self._kind.clear()

    def armed(self, point: str) -> bool:
        with self._lock:
            return point in self._armed

    def hit(self, point: str) -> None:
        with self._lock:
            if point not in self._armed:
                return
            if self._armed[point] > 0:
                self._armed[point] -= 1
                return
            kind = self._kind.pop(point)
            self._armed.pop(point)
        if kind == "exit":
            os._exit(42)
        raise InjectedFailure(f"injected failure at {point}")


FAULTS = _Faults()
