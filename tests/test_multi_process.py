"""SharedLock/SharedQueue/SharedDict/SharedMemory across real processes."""

import multiprocessing as mp
import queue as pyqueue
import time

import numpy as np
import pytest

from dlrover_amd.common.multi_process import (
    IPCServer,
    SharedDict,
    SharedLock,
    SharedQueue,
    attach_shared_memory,
    create_shared_memory,
    unlink_shared_memory,
)


@pytest.fixture()
def ipc_server(tmp_path):
    server = IPCServer(socket_path=str(tmp_path / "t.sock")).start()
    yield server
    server.stop()


def test_shared_lock(ipc_server):
    path = ipc_server.socket_path
    a = SharedLock("ckpt", socket_path=path)
    b = SharedLock("ckpt", socket_path=path)
    assert a.acquire()
    assert not b.acquire(blocking=False)
    assert a.locked() and b.locked()
    a.release()
    assert b.acquire(blocking=False)
    b.release()


def test_shared_queue(ipc_server):
    path = ipc_server.socket_path
    q1 = SharedQueue("events", socket_path=path)
    q2 = SharedQueue("events", socket_path=path)
    q1.put({"type": "SAVE", "step": 10})
    assert q2.qsize() == 1
    assert q2.get(timeout=2) == {"type": "SAVE", "step": 10}
    assert q2.empty()
    with pytest.raises(pyqueue.Empty):
        q2.get(block=False)


def test_shared_dict(ipc_server):
    path = ipc_server.socket_path
    d1 = SharedDict("meta", socket_path=path)
    d2 = SharedDict("meta", socket_path=path)
    d1.set("step", 5)
    d1.update({"shapes": [1, 2, 3]})
    assert d2.get("step") == 5
    assert d2.get_all() == {"step": 5, "shapes": [1, 2, 3]}
    d2.delete("step")
    assert d1.get("step") is None


def _child_lock_holder(path, held_evt, release_evt):
    lock = SharedLock("cross", socket_path=path)
    lock.acquire()
    held_evt.set()
    release_evt.wait(timeout=10)
    lock.release()


def test_lock_across_processes(ipc_server):
    path = ipc_server.socket_path
    ctx = mp.get_context("spawn")
    held, release = ctx.Event(), ctx.Event()
    p = ctx.Process(target=_child_lock_holder, args=(path, held, release))
    p.start()
    try:
        assert held.wait(timeout=15)
        mine = SharedLock("cross", socket_path=path)
        assert not mine.acquire(blocking=False)
        release.set()
        deadline = time.time() + 10
        got = False
        while time.time() < deadline:
            if mine.acquire(blocking=False):
                got = True
                break
            time.sleep(0.05)
        assert got
        mine.release()
    finally:
        release.set()
        p.join(timeout=10)


def test_shared_memory_roundtrip():
    name = f"dlrover_test_{time.time_ns()}"
    shm = create_shared_memory(name, 1024)
    try:
        arr = np.ndarray((256,), dtype=np.float32, buffer=shm.buf)
        arr[:] = np.arange(256, dtype=np.float32)
        peer = attach_shared_memory(name)
        assert peer is not None
        arr2 = np.ndarray((256,), dtype=np.float32, buffer=peer.buf)
        assert np.array_equal(arr2, np.arange(256, dtype=np.float32))
        peer.close()
    finally:
        shm.close()
        unlink_shared_memory(name)


def test_shared_memory_grow():
    name = f"dlrover_grow_{time.time_ns()}"
    shm = create_shared_memory(name, 128)
    shm.close()
    shm = create_shared_memory(name, 4096)
    try:
        assert shm.size >= 4096
    finally:
        shm.close()
        unlink_shared_memory(name)
