"""DAG core tests — run against whichever implementation is active
(C++ _core when built, python fallback otherwise); the interface is
identical (reference analogue: graph-executor-2 Algorithms tests)."""
import pytest

from lzy_amd.sched import Dag, Journal, xxhash64


def test_linear_chain():
    d = Dag()
    d.add_task("a", [])
    d.add_task("b", ["a"])
    d.add_task("c", ["b"])
    d.seal()
    assert d.take_ready() == ["a"]
    assert d.take_ready() == []
    assert d.complete("a") == ["b"]
    assert d.complete("b") == ["c"]
    assert d.complete("c") == []
    assert d.is_done()


def test_diamond():
    d = Dag()
    d.add_task("src", [])
    d.add_task("l", ["src"])
    d.add_task("r", ["src"])
    d.add_task("sink", ["l", "r"])
    d.seal()
    assert d.take_ready() == ["src"]
    newly = d.complete("src")
    assert sorted(newly) == ["l", "r"]
    assert d.complete("l") == []
    assert d.complete("r") == ["sink"]
    d.complete("sink")
    assert d.is_done()


def test_failure_cancels_downstream():
    d = Dag()
    d.add_task("a", [])
    d.add_task("b", ["a"])
    d.add_task("c", ["b"])
    d.add_task("x", [])  # independent
    d.seal()
    ready = d.take_ready()
    assert sorted(ready) == ["a", "x"]
    cancelled = d.fail("a")
    assert sorted(cancelled) == ["b", "c"]
    d.complete("x")
    assert d.is_done()


def test_unknown_deps_dropped():
    # deps on entries produced before this batch are not graph edges
    d = Dag()
    d.add_task("a", ["not-in-graph"])
    d.seal()
    assert d.take_ready() == ["a"]


def test_cycle_detected():
    d = Dag()
    d.add_task("a", ["b"])
    d.add_task("b", ["a"])
    with pytest.raises(Exception):
        d.seal()


def test_duplicate_task_rejected():
    d = Dag()
    d.add_task("a", [])
    with pytest.raises(Exception):
        d.add_task("a", [])


def test_wide_fanout():
    d = Dag()
    d.add_task("root", [])
    for i in range(100):
        d.add_task(f"t{i}", ["root"])
    d.add_task("sink", [f"t{i}" for i in range(100)])
    d.seal()
    assert d.take_ready() == ["root"]
    newly = d.complete("root")
    assert len(newly) == 100
    for i in range(99):
        assert d.complete(f"t{i}") == []
    assert d.complete("t99") == ["sink"]


def test_journal_replay(tmp_path):
    path = str(tmp_path / "j.jsonl")
    j = Journal(path)
    j.record("t1", "scheduled")
    j.record("t1", "running")
    j.record("t1", "done")
    j.record("t2", "running")
    j.close()
    states = Journal.replay(path)
    assert states == {"t1": "done", "t2": "running"}


def test_journal_replay_torn_tail(tmp_path):
    path = str(tmp_path / "j.jsonl")
    j = Journal(path)
    j.record("t1", "done")
    j.close()
    with open(path, "a") as f:
        f.write('{"t": "t2", "s": "runn')  # crash mid-write
    states = Journal.replay(path)
    assert states == {"t1": "done"}


def test_xxhash64_deterministic():
    a = xxhash64(b"hello world")
    b = xxhash64(b"hello world")
    c = xxhash64(b"hello worlds")
    assert a == b
    assert a != c
    assert isinstance(a, int)


def test_xxhash64_known_vectors():
    # official XXH64 test vectors (seed 0)
    assert xxhash64(b"") == 0xEF46DB3751D8E999
    assert xxhash64(b"hello") == 0x26C7827D889F6DA3
