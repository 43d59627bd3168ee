"""Pure-python fallback for the C++ DAG core (same interface).

Semantics mirror the native core: frontier scheduling with transitive
cancellation on failure (reference analogue: graph-executor-2
algo/Algorithms.java:10 buildTaskDependents + cycle check :76).
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Set


class PyDag:
    def __init__(self) -> None:
        self._deps: Dict[str, Set[str]] = {}
        self._dependents: Dict[str, List[str]] = {}
        self._indegree: Dict[str, int] = {}
        self._state: Dict[str, str] = {}  # pending|ready|running|done|failed|cancelled
        self._sealed = False

    def add_task(self, tid: str, deps: List[str]) -> None:
        if self._sealed:
            raise RuntimeError("DAG is sealed")
        if tid in self._deps:
            raise ValueError(f"duplicate task {tid}")
        self._deps[tid] = set(deps)
        self._state[tid] = "pending"

    def seal(self) -> None:
        # drop deps on unknown tasks (entries produced before this batch)
        for tid in self._deps:
            self._deps[tid] = {d for d in self._deps[tid] if d in self._deps}
        self._check_acyclic()
        for tid, deps in self._deps.items():
            self._indegree[tid] = len(deps)
            for d in deps:
                self._dependents.setdefault(d, []).append(tid)
            if not deps:
                self._state[tid] = "ready"
        self._sealed = True

    def _check_acyclic(self) -> None:
        indeg = {t: len(d) for t, d in self._deps.items()}
        frontier = [t for t, n in indeg.items() if n == 0]
        seen = 0
        while frontier:
            t = frontier.pop()
            seen += 1
            for dep in self._dependents_of(t):
                indeg[dep] -= 1
                if indeg[dep] == 0:
                    frontier.append(dep)
        if seen != len(self._deps):
            raise ValueError("cycle detected in task graph")

    def _dependents_of(self, tid: str) -> List[str]:
        return [t for t, d in self._deps.items() if tid in d]

    def take_ready(self) -> List[str]:
        out = [t for t, s in self._state.items() if s == "ready"]
        for t in out:
            self._state[t] = "running"
        return out

    def complete(self, tid: str) -> List[str]:
        self._state[tid] = "done"
        newly = []
        for dep in self._dependents.get(tid, []):
            self._indegree[dep] -= 1
            if self._indegree[dep] == 0 and self._state[dep] == "pending":
                self._state[dep] = "ready"
                newly.append(dep)
        return newly

    def fail(self, tid: str) -> List[str]:
        self._state[tid] = "failed"
        cancelled: List[str] = []
        stack = list(self._dependents.get(tid, []))
        while stack:
            t = stack.pop()
            if self._state[t] in ("pending", "ready"):
                self._state[t] = "cancelled"
                cancelled.append(t)
                stack.extend(self._dependents.get(t, []))
        return cancelled

    def state(self, tid: str) -> str:
        return self._state[tid]

    def is_done(self) -> bool:
        return all(s in ("done", "failed", "cancelled") for s in self._state.values())

    def counts(self) -> Dict[str, int]:
        out: Dict[str, int] = {}
        for s in self._state.values():
            out[s] = out.get(s, 0) + 1
        return out


class PyJournal:
    """Append-only task-state journal for crash resume (reference analogue:
    the durable LRO step machine, long-running/OperationRunnerBase.java:27)."""

    def __init__(self, path: str, sync: bool = False) -> None:
        self._path = path
        self._sync = sync
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        self._f = open(path, "a", buffering=1)

    def record(self, tid: str, state: str, detail: str = "") -> None:
        self._f.write(json.dumps({"t": tid, "s": state, "d": detail}) + "\n")
        self._f.flush()
        if self._sync:
            os.fsync(self._f.fileno())

    def sync(self) -> None:
        os.fsync(self._f.fileno())

    def close(self) -> None:
        if not self._f.closed:
            self._f.flush()
            os.fsync(self._f.fileno())
            self._f.close()

    @staticmethod
    def replay(path: str) -> Dict[str, str]:
        states: Dict[str, str] = {}
        if not os.path.exists(path):
            return states
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                try:
                    rec = json.loads(line)
                except json.JSONDecodeError:
                    continue  # torn tail write after a crash
                states[rec["t"]] = rec["s"]
        return states
