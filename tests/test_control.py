"""Control-plane unit tests: AF_UNIX star (driver <-> workers) without
torch.distributed — connection handshake, command/event flow, broadcast,
and death detection (reference: the gRPC control mesh + allocator
heartbeats)."""
import threading
import time

import pytest

from lzy_amd.channels.control import DriverControl, WorkerControl


def _mk(world, events):
    drv = DriverControl(world, on_event=lambda r, m: events.append((r, m)))
    workers = {}

    def connect(rank):
        workers[rank] = WorkerControl(rank, drv.address)

    ths = [threading.Thread(target=connect, args=(r,)) for r in range(world)]
    for t in ths:
        t.start()
    drv.accept_all()
    for t in ths:
        t.join()
    return drv, workers


def test_command_and_event_roundtrip():
    events = []
    drv, workers = _mk(2, events)
    try:
        drv.send(1, {"cmd": "ping", "x": 7})
        msg = workers[1].recv()
        assert msg == {"cmd": "ping", "x": 7}
        workers[1].send_event({"ev": "pong", "x": 8})
        for _ in range(100):
            if events:
                break
            time.sleep(0.01)
        assert (1, {"ev": "pong", "x": 8}) in events
    finally:
        drv.close()


def test_broadcast_reaches_all():
    events = []
    drv, workers = _mk(3, events)
    try:
        drv.broadcast({"cmd": "hello"})
        for r in range(3):
            assert workers[r].recv() == {"cmd": "hello"}
    finally:
        drv.close()


def test_worker_death_emits_worker_lost():
    events = []
    drv, workers = _mk(2, events)
    try:
        workers[1].close()
        for _ in range(200):
            if any(m.get("ev") == "worker_lost" and r == 1 for r, m in events):
                break
            time.sleep(0.01)
        assert any(m.get("ev") == "worker_lost" and r == 1 for r, m in events)
    finally:
        drv.close()


def test_version_gate_rejects_mismatch():
    from multiprocessing.connection import Client

    from lzy_amd.channels.control import _AUTHKEY

    drv = DriverControl(1, on_event=lambda r, m: None)

    def connect():
        try:
            conn = Client(drv.address, family="AF_UNIX", authkey=_AUTHKEY)
            conn.send({"rank": 0, "version": "0.0.0-other", "torch": "x"})
            time.sleep(0.5)
            conn.close()
        except Exception:
            pass

    t = threading.Thread(target=connect)
    t.start()
    with pytest.raises(RuntimeError, match="version mismatch"):
        drv.accept_all()
    t.join()
    drv.close()
