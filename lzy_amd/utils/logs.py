"""Per-op std-log capture with live tail.

Reference capability: worker stdout/stderr flow to a Kafka topic per
execution and stream back to the client with ``[LZY-REMOTE-<task>]``
prefixes (reference: lzy/worker KafkaLogsWriter + pylzy runtime.py:283-301).
In-process re-design: ops run in worker threads/processes of the same node,
so "log streaming" is a thread-routing tee — each op thread's writes are
prefixed and forwarded to the real stream immediately (live tail), and kept
in a per-task buffer for post-mortem retrieval.
"""
from __future__ import annotations

import io
import sys
import threading
from typing import Dict, Optional, TextIO, Tuple


class _ThreadRoutingTee(io.TextIOBase):
    def __init__(self, fallback: TextIO):
        self._fallback = fallback
        self._routes: Dict[int, Tuple[str, io.StringIO]] = {}
        self._lock = threading.Lock()

    def route(self, prefix: str, echo: bool = True) -> io.StringIO:
        buf = io.StringIO()
        with self._lock:
            self._routes[threading.get_ident()] = (prefix, buf, echo)
        return buf

    def unroute(self) -> None:
        with self._lock:
            self._routes.pop(threading.get_ident(), None)

    def write(self, s: str) -> int:
        ident = threading.get_ident()
        with self._lock:
            entry = self._routes.get(ident)
        if entry is None:
            return self._fallback.write(s)
        prefix, buf, echo = entry
        buf.write(s)
        if echo:
            for line in s.splitlines(keepends=True):
                self._fallback.write(f"{prefix}{line}" if line.strip() else line)
        return len(s)

    def flush(self) -> None:
        self._fallback.flush()

    @property
    def encoding(self):  # type: ignore[override]
        return getattr(self._fallback, "encoding", "utf-8")

    def isatty(self) -> bool:
        return False


class OpLogCapture:
    """Installs routing tees over sys.stdout/sys.stderr while active."""

    _installed: Optional["OpLogCapture"] = None
    _lock = threading.Lock()

    def __init__(self) -> None:
        self._out_tee: Optional[_ThreadRoutingTee] = None
        self._err_tee: Optional[_ThreadRoutingTee] = None
        self._depth = 0

    @classmethod
    def instance(cls) -> "OpLogCapture":
        with cls._lock:
            if cls._installed is None:
                cls._installed = cls()
            return cls._installed

    def install(self) -> None:
        with self._lock:
            self._depth += 1
            if self._depth == 1:
                self._out_tee = _ThreadRoutingTee(sys.stdout)
                self._err_tee = _ThreadRoutingTee(sys.stderr)
                sys.stdout = self._out_tee  # type: ignore[assignment]
                sys.stderr = self._err_tee  # type: ignore[assignment]

    def uninstall(self) -> None:
        with self._lock:
            if self._depth == 0:
                return  # already uninstalled (double teardown is benign)
            self._depth -= 1
            if self._depth == 0 and self._out_tee is not None:
                sys.stdout = self._out_tee._fallback  # type: ignore[assignment]
                sys.stderr = self._err_tee._fallback  # type: ignore[assignment]
                self._out_tee = None
                self._err_tee = None

    def route_current_thread(
        self, task_name: str, echo: bool = True
    ) -> Tuple[io.StringIO, io.StringIO]:
        """``echo=False``: capture only — pool workers on ranks > 0 do not
        write to their own console; their logs travel to the driver in the
        TaskResult and are tailed there (reference: worker→Kafka→client
        ReadStdSlots stream)."""
        prefix = f"[LZY-{task_name}] "
        out = self._out_tee.route(prefix, echo) if self._out_tee else io.StringIO()
        err = self._err_tee.route(prefix, echo) if self._err_tee else io.StringIO()
        return out, err

    def unroute_current_thread(self) -> None:
        if self._out_tee:
            self._out_tee.unroute()
        if self._err_tee:
            self._err_tee.unroute()
