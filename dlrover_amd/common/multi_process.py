"""Agent↔trainer IPC: named locks, queues and dicts over a unix socket, plus
POSIX shared memory.

Parity target: ref dlrover/python/common/multi_process.py:38-747 (SharedLock
:263, SharedQueue :455, SharedDict :579, SharedMemory :675). Redesigned: the
reference runs one socket server per primitive; here a single ``IPCServer``
in the agent process hosts a registry of named primitives and clients address
them by (kind, name) — fewer fds, one accept loop, same semantics.

The flash-checkpoint engine in the *training* process uses these to
coordinate with the AsyncCheckpointSaver in the *agent* process:
SharedLock serializes shm writes against persistence, SharedQueue carries
save events, SharedDict carries tensor metadata.
"""

import os
import pickle
import queue
import socket
import socketserver
import threading
import time
from multiprocessing import shared_memory
from typing import Dict, Optional

from dlrover_amd.common.log import logger

SOCKET_DIR_ENV = "DLROVER_IPC_SOCKET_DIR"
_DEF_DIR = "/tmp/dlrover_amd_ipc"


def ipc_socket_path(job_name: str = "") -> str:
    base = os.getenv(SOCKET_DIR_ENV, _DEF_DIR)
    job = job_name or os.getenv("ELASTIC_JOB_NAME", "default")
    d = os.path.join(base, job)
    os.makedirs(d, exist_ok=True)
    try:
        os.chmod(d, 0o700)  # raw pickle flows over this socket: owner-only
    except OSError:
        pass
    return os.path.join(d, "ipc.sock")


def _send_msg(sock: socket.socket, obj) -> None:
    data = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
    sock.sendall(len(data).to_bytes(8, "little") + data)


def _recv_msg(sock: socket.socket):
    head = b""
    while len(head) < 8:
        chunk = sock.recv(8 - len(head))
        if not chunk:
            raise ConnectionError("IPC peer closed")
        head += chunk
    size = int.from_bytes(head, "little")
    buf = bytearray()
    while len(buf) < size:
        chunk = sock.recv(min(1 << 20, size - len(buf)))
        if not chunk:
            raise ConnectionError("IPC peer closed mid-message")
        buf += chunk
    return pickle.loads(bytes(buf))


class _Handler(socketserver.BaseRequestHandler):
    def handle(self):
        server: "IPCServer" = self.server.ipc_server  # type: ignore[attr-defined]
        while True:
            try:
                req = _recv_msg(self.request)
            except (ConnectionError, OSError):
                return
            try:
                resp = server.dispatch(req, self.request)
            except Exception as e:  # noqa: BLE001 — report errors to client
                resp = {"ok": False, "error": repr(e)}
            try:
                _send_msg(self.request, resp)
            except (ConnectionError, OSError):
                return


class _ThreadedUnixServer(socketserver.ThreadingUnixStreamServer):
    daemon_threads = True
    allow_reuse_address = True


class IPCServer:
    """Hosts all named IPC primitives for one agent process."""

    def __init__(self, socket_path: Optional[str] = None):
        self._path = socket_path or ipc_socket_path()
        if os.path.exists(self._path):
            os.unlink(self._path)
        self._locks: Dict[str, dict] = {}
        self._queues: Dict[str, queue.Queue] = {}
        self._dicts: Dict[str, dict] = {}
        self._meta_lock = threading.Lock()
        self._server = _ThreadedUnixServer(self._path, _Handler)
        self._server.ipc_server = self  # type: ignore[attr-defined]
        self._thread = threading.Thread(
            target=self._server.serve_forever, name="dlrover-ipc", daemon=True
        )

    @property
    def socket_path(self) -> str:
        return self._path

    def start(self):
        self._thread.start()
        logger.info("IPCServer listening on %s", self._path)
        return self

    def stop(self):
        self._server.shutdown()
        self._server.server_close()
        if os.path.exists(self._path):
            try:
                os.unlink(self._path)
            except OSError:
                pass

    # -- primitive registry ---------------------------------------------------

    def _lock_obj(self, name: str) -> dict:
        with self._meta_lock:
            return self._locks.setdefault(
                name, {"lock": threading.Lock(), "owner": None}
            )

    def _queue_obj(self, name: str, maxsize: int = 0) -> queue.Queue:
        with self._meta_lock:
            if name not in self._queues:
                self._queues[name] = queue.Queue(maxsize=maxsize)
            return self._queues[name]

    def _dict_obj(self, name: str) -> dict:
        with self._meta_lock:
            return self._dicts.setdefault(name, {})

    # -- dispatch -------------------------------------------------------------

    def dispatch(self, req: dict, conn) -> dict:
        kind, op, name = req["kind"], req["op"], req["name"]
        if kind == "lock":
            return self._handle_lock(op, name, req)
        if kind == "queue":
            return self._handle_queue(op, name, req)
        if kind == "dict":
            return self._handle_dict(op, name, req)
        if kind == "ping":
            return {"ok": True}
        raise ValueError(f"unknown IPC kind {kind!r}")

    def _handle_lock(self, op, name, req):
        obj = self._lock_obj(name)
        if op == "acquire":
            ok = obj["lock"].acquire(
                blocking=req.get("blocking", True),
                timeout=req.get("timeout", -1) if req.get("blocking", True) else -1,
            )
            if ok:
                obj["owner"] = req.get("owner")
            return {"ok": True, "result": ok}
        if op == "release":
            try:
                obj["lock"].release()
                obj["owner"] = None
                return {"ok": True, "result": True}
            except RuntimeError:
                return {"ok": True, "result": False}
        if op == "locked":
            return {"ok": True, "result": obj["lock"].locked()}
        raise ValueError(f"unknown lock op {op!r}")

    def _handle_queue(self, op, name, req):
        q = self._queue_obj(name, req.get("maxsize", 0))
        if op == "put":
            q.put(req["value"], block=req.get("block", True), timeout=req.get("timeout"))
            return {"ok": True}
        if op == "get":
            try:
                v = q.get(block=req.get("block", True), timeout=req.get("timeout"))
                return {"ok": True, "result": v, "empty": False}
            except queue.Empty:
                return {"ok": True, "empty": True}
        if op == "qsize":
            return {"ok": True, "result": q.qsize()}
        if op == "empty":
            return {"ok": True, "result": q.empty()}
        raise ValueError(f"unknown queue op {op!r}")

    def _handle_dict(self, op, name, req):
        d = self._dict_obj(name)
        if op == "set":
            d[req["key"]] = req["value"]
            return {"ok": True}
        if op == "update":
            d.update(req["value"])
            return {"ok": True}
        if op == "get":
            return {"ok": True, "result": d.get(req["key"], req.get("default"))}
        if op == "getall":
            return {"ok": True, "result": dict(d)}
        if op == "delete":
            d.pop(req["key"], None)
            return {"ok": True}
        if op == "clear":
            d.clear()
            return {"ok": True}
        raise ValueError(f"unknown dict op {op!r}")


class _IPCClient:
    """One connection to the agent's IPCServer; thread-safe via a lock."""

    def __init__(self, socket_path: Optional[str] = None, connect_timeout: float = 60.0):
        self._path = socket_path or ipc_socket_path()
        self._sock: Optional[socket.socket] = None
        self._lock = threading.Lock()
        self._connect_timeout = connect_timeout

    def _ensure(self):
        if self._sock is None:
            deadline = time.time() + self._connect_timeout
            while True:
                try:
                    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
                    s.connect(self._path)
                    self._sock = s
                    return
                except (FileNotFoundError, ConnectionRefusedError):
                    if time.time() > deadline:
                        raise TimeoutError(
                            f"cannot reach IPCServer at {self._path}"
                        ) from None
                    time.sleep(0.2)

    def call(self, req: dict):
        with self._lock:
            self._ensure()
            try:
                _send_msg(self._sock, req)
                resp = _recv_msg(self._sock)
            except (ConnectionError, OSError):
                # reconnect once (agent may have restarted its server)
                self._sock = None
                self._ensure()
                _send_msg(self._sock, req)
                resp = _recv_msg(self._sock)
        if not resp.get("ok", False):
            raise RuntimeError(f"IPC error: {resp.get('error')}")
        return resp

    def close(self):
        with self._lock:
            if self._sock is not None:
                try:
                    self._sock.close()
                finally:
                    self._sock = None


class SharedLock:
    """Named lock served by the agent (ref: multi_process.py:263)."""

    def __init__(self, name: str, socket_path: Optional[str] = None):
        self.name = name
        self._client = _IPCClient(socket_path)
        self._owner = f"{os.getpid()}-{threading.get_ident()}"

    def acquire(self, blocking: bool = True, timeout: float = -1) -> bool:
        r = self._client.call(
            {
                "kind": "lock",
                "op": "acquire",
                "name": self.name,
                "blocking": blocking,
                "timeout": timeout,
                "owner": self._owner,
            }
        )
        return bool(r["result"])

    def release(self) -> bool:
        r = self._client.call({"kind": "lock", "op": "release", "name": self.name})
        return bool(r["result"])

    def locked(self) -> bool:
        r = self._client.call({"kind": "lock", "op": "locked", "name": self.name})
        return bool(r["result"])

    def __enter__(self):
        self.acquire()
        return self

    def __exit__(self, *exc):
        self.release()


class SharedQueue:
    """Named queue served by the agent (ref: multi_process.py:455)."""

    def __init__(self, name: str, maxsize: int = 0, socket_path: Optional[str] = None):
        self.name = name
        self.maxsize = maxsize
        self._client = _IPCClient(socket_path)

    def put(self, value, block: bool = True, timeout: Optional[float] = None):
        self._client.call(
            {
                "kind": "queue",
                "op": "put",
                "name": self.name,
                "maxsize": self.maxsize,
                "value": value,
                "block": block,
                "timeout": timeout,
            }
        )

    def get(self, block: bool = True, timeout: Optional[float] = None):
        if block and timeout is None:
            # poll so a dead server raises instead of hanging forever
            while True:
                r = self._client.call(
                    {
                        "kind": "queue",
                        "op": "get",
                        "name": self.name,
                        "maxsize": self.maxsize,
                        "block": True,
                        "timeout": 5.0,
                    }
                )
                if not r.get("empty", True):
                    return r["result"]
        r = self._client.call(
            {
                "kind": "queue",
                "op": "get",
                "name": self.name,
                "maxsize": self.maxsize,
                "block": block,
                "timeout": timeout,
            }
        )
        if r.get("empty", True):
            raise queue.Empty
        return r["result"]

    def qsize(self) -> int:
        return self._client.call({"kind": "queue", "op": "qsize", "name": self.name})["result"]

    def empty(self) -> bool:
        return self._client.call({"kind": "queue", "op": "empty", "name": self.name})["result"]


class SharedDict:
    """Named dict served by the agent (ref: multi_process.py:579)."""

    def __init__(self, name: str, socket_path: Optional[str] = None):
        self.name = name
        self._client = _IPCClient(socket_path)

    def set(self, key, value):
        self._client.call(
            {"kind": "dict", "op": "set", "name": self.name, "key": key, "value": value}
        )

    def update(self, mapping: dict):
        self._client.call(
            {"kind": "dict", "op": "update", "name": self.name, "value": mapping}
        )

    def get(self, key, default=None):
        return self._client.call(
            {"kind": "dict", "op": "get", "name": self.name, "key": key, "default": default}
        )["result"]

    def get_all(self) -> dict:
        return self._client.call({"kind": "dict", "op": "getall", "name": self.name})["result"]

    def delete(self, key):
        self._client.call({"kind": "dict", "op": "delete", "name": self.name, "key": key})

    def clear(self):
        self._client.call({"kind": "dict", "op": "clear", "name": self.name})


# ---------------------------------------------------------------------------
# POSIX shared memory without resource-tracker interference
# ---------------------------------------------------------------------------


def _untrack(shm: shared_memory.SharedMemory):
    """Python's resource_tracker unlinks shm segments when *any* process that
    touched them exits — wrong for our cross-process lifetime (the agent owns
    the segment; trainers come and go). Same workaround as the reference
    (ref: multi_process.py:675 SharedMemory subclass)."""
    try:
        from multiprocessing import resource_tracker

        resource_tracker.unregister(shm._name, "shared_memory")  # type: ignore[attr-defined]
    except Exception:  # noqa: BLE001 — best effort, py-version dependent
        pass


def create_shared_memory(name: str, size: int) -> shared_memory.SharedMemory:
    """Create (or replace) a named shm segment of at least ``size`` bytes."""
    try:
        old = shared_memory.SharedMemory(name=name)
        if old.size >= size:
            _untrack(old)
            return old
        old.close()
        old.unlink()
    except FileNotFoundError:
        pass
    shm = shared_memory.SharedMemory(name=name, create=True, size=size)
    _untrack(shm)
    return shm


def attach_shared_memory(name: str) -> Optional[shared_memory.SharedMemory]:
    try:
        shm = shared_memory.SharedMemory(name=name)
        _untrack(shm)
        return shm
    except FileNotFoundError:
        return None


def unlink_shared_memory(name: str):
    try:
        shm = shared_memory.SharedMemory(name=name)
        shm.close()
        shm.unlink()
    except FileNotFoundError:
        pass
