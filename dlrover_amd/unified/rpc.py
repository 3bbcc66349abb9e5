"""Per-actor RPC for the unified architecture.

Parity target: ref dlrover/python/unified/api/runtime/rpc_helper.py:334 —
the reference exposes named methods on Ray actors and calls them
cross-role (e.g. the trainer pulling rollouts, the controller poking
workers). Local backend: one multiprocessing Listener per actor; the
master (or any peer) connects by actor name through an address registry
on disk (works across spawn'd processes).
"""

import os
import pickle
import threading
from multiprocessing.connection import Client, Listener
from typing import Any, Callable, Dict, Optional

from dlrover_amd.common.log import logger

_AUTH = b"dlrover-amd-rpc"


def _registry_dir(job: str) -> str:
    d = os.path.join(
        os.getenv("DLROVER_IPC_SOCKET_DIR", "/tmp"), f"prime_rpc_{job}"
    )
    os.makedirs(d, exist_ok=True)
    return d


class ActorRpcServer:
    """Worker-side: export named handlers; one background thread serves."""

    def __init__(self, job: str, actor_name: str):
        self.job = job
        self.actor_name = actor_name
        self._handlers: Dict[str, Callable] = {}
        addr = os.path.join(_registry_dir(job), actor_name + ".sock")
        try:
            os.unlink(addr)
        except OSError:
            pass
        self._listener = Listener(addr, "AF_UNIX", authkey=_AUTH)
        self.address = addr
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def register(self, name: str, fn: Callable) -> "ActorRpcServer":
        self._handlers[name] = fn
        return self

    def start(self) -> "ActorRpcServer":
        self._thread = threading.Thread(
            target=self._serve, name=f"rpc-{self.actor_name}", daemon=True
        )
        self._thread.start()
        return self

    def _serve(self):
        while not self._stop.is_set():
            try:
                conn = self._listener.accept()
            except OSError:
                return
            try:
                method, args, kwargs = conn.recv()
                fn = self._handlers.get(method)
                if fn is None:
                    conn.send(("error", f"no such method {method}"))
                else:
                    try:
                        conn.send(("ok", fn(*args, **kwargs)))
                    except Exception as e:  # noqa: BLE001 — marshal to caller
                        conn.send(("error", repr(e)))
            except (EOFError, pickle.PickleError, OSError) as e:
                logger.warning("rpc %s: bad request: %s", self.actor_name, e)
            finally:
                conn.close()

    def stop(self):
        self._stop.set()
        try:
            self._listener.close()
        except OSError:
            pass


def call_actor(job: str, actor_name: str, method: str, *args,
               timeout: float = 30.0, **kwargs) -> Any:
    """Client side: one call = one connection (simple, restart-tolerant)."""
    addr = os.path.join(_registry_dir(job), actor_name + ".sock")
    conn = Client(addr, "AF_UNIX", authkey=_AUTH)
    try:
        conn.send((method, args, kwargs))
        if not conn.poll(timeout):
            raise TimeoutError(f"rpc {actor_name}.{method} timed out")
        status, payload = conn.recv()
    finally:
        conn.close()
    if status != "ok":
        raise RuntimeError(f"rpc {actor_name}.{method}: {payload}")
    return payload


def serve_current_actor(handlers: Dict[str, Callable]) -> ActorRpcServer:
    """Inside a PrimeMaster-launched worker: export handlers under this
    vertex's name (ROLE-RANK, from the env the launcher set)."""
    job = os.getenv("DLROVER_PRIME_JOB", "dljob")
    name = f"{os.getenv('ROLE', 'worker')}-{os.getenv('RANK', '0')}"
    srv = ActorRpcServer(job, name)
    for k, v in handlers.items():
        srv.register(k, v)
    return srv.start()
