"""Metrics: counters/gauges/histograms + Prometheus text exposition.

Reference capability (util-log/metrics/PrometheusMetricReporter.java:7 and
per-service metrics like AllocatorMetrics): Prometheus counters per
subsystem served over HTTP.  Single-node re-design: a process-local
registry with the standard text format, servable via a tiny HTTP endpoint
(``serve()``) or dumpable to a string; per-op timings feed the scheduler's
overhead accounting that bench.py reports.
"""
from __future__ import annotations

import threading
import time
from collections import defaultdict
from typing import Dict, List, Optional, Tuple


class Metrics:
    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._counters: Dict[Tuple[str, Tuple[Tuple[str, str], ...]], float] = defaultdict(float)
        self._gauges: Dict[Tuple[str, Tuple[Tuple[str, str], ...]], float] = {}
        self._timings: Dict[str, List[float]] = defaultdict(list)
        self._server = None

    def inc(self, name: str, value: float = 1.0, **labels: str) -> None:
        with self._lock:
            self._counters[(name, tuple(sorted(labels.items())))] += value

    def counter_value(self, name: str, **labels: str) -> float:
        with self._lock:
            return self._counters.get((name, tuple(sorted(labels.items()))), 0.0)

    def set_gauge(self, name: str, value: float, **labels: str) -> None:
        with self._lock:
            self._gauges[(name, tuple(sorted(labels.items())))] = value

    # per-series cap: long-lived services observe forever; an unbounded
    # list is a slow memory leak (~32 floats per workflow in the pool).
    # When full, the series halves to its most recent half — stats
    # become a recent-window view, which is what dashboards want anyway.
    TIMING_CAP = 200_000

    def observe(self, name: str, seconds: float) -> None:
        with self._lock:
            xs = self._timings[name]
            xs.append(seconds)
            if len(xs) > self.TIMING_CAP:
                del xs[: len(xs) // 2]

    def timing_stats(self, name: str) -> Dict[str, float]:
        with self._lock:
            xs = sorted(self._timings.get(name, []))
        if not xs:
            return {"count": 0}
        n = len(xs)
        return {
            "count": n,
            "mean": sum(xs) / n,
            "p50": xs[n // 2],
            "p99": xs[min(n - 1, int(n * 0.99))],
            "max": xs[-1],
            "total": sum(xs),
        }

    def reset(self) -> None:
        with self._lock:
            self._counters.clear()
            self._gauges.clear()
            self._timings.clear()

    # -- exposition ---------------------------------------------------------

    @staticmethod
    def _fmt_labels(labels: Tuple[Tuple[str, str], ...]) -> str:
        if not labels:
            return ""
        inner = ",".join(f'{k}="{v}"' for k, v in labels)
        return "{" + inner + "}"

    def render(self) -> str:
        lines: List[str] = []
        with self._lock:
            for (name, labels), v in sorted(self._counters.items()):
                lines.append(f"{name}{self._fmt_labels(labels)} {v}")
            for (name, labels), v in sorted(self._gauges.items()):
                lines.append(f"{name}{self._fmt_labels(labels)} {v}")
            for name, xs in sorted(self._timings.items()):
                if xs:
                    lines.append(f"{name}_seconds_count {len(xs)}")
                    lines.append(f"{name}_seconds_sum {sum(xs)}")
        return "\n".join(lines) + "\n"

    def serve(self, port: int = 0) -> int:
        """Start a /metrics HTTP endpoint; returns the bound port."""
        import http.server
        import socketserver

        metrics = self

        class Handler(http.server.BaseHTTPRequestHandler):
            def do_GET(self):  # noqa: N802
                body = metrics.render().encode()
                self.send_response(200)
                self.send_header("Content-Type", "text/plain; version=0.0.4")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *a):
                pass

        srv = socketserver.TCPServer(("127.0.0.1", port), Handler)
        t = threading.Thread(target=srv.serve_forever, daemon=True)
        t.start()
        self._server = srv
        return srv.server_address[1]

    def stop(self) -> None:
        if self._server is not None:
            self._server.shutdown()
            self._server = None


METRICS = Metrics()


class timed:
    """Context manager: METRICS.observe(name, elapsed)."""

    def __init__(self, name: str):
        self.name = name

    def __enter__(self):
        self.t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        METRICS.observe(self.name, time.perf_counter() - self.t0)
        return False
