"""Agent-side async checkpoint persistence.

Parity target: ref dlrover/python/elastic_agent/torch/ckpt_saver.py:399-1494
(AsyncCheckpointSaver: factory thread, _sync_shm_to_storage event loop,
signal-handler persist on SIGTERM, done-file + tracker-file two-phase commit).

Runs in the AGENT process. Training processes write shm snapshots
(flash_checkpoint.shm_handler) and enqueue CheckpointEvents on the shared
queue; this saver drains the queue, serializes each local rank's shm segment
to ``<path>/rank_<global_rank>.pt``, drops a ``.done_<rank>`` marker, and
commits by writing ``dlrover_latest.txt`` once every expected shard's marker
exists (works across nodes on shared storage). On worker failure or SIGTERM
the agent calls save_shm_to_storage() to persist whatever committed snapshot
the shm still holds — the crash-consistency path the goodput metric depends
on.
"""

import os
import queue as pyqueue
import signal
import threading
import time
from typing import Dict, List, Optional

import torch

from dlrover_amd.common.constants import CheckpointConstant
from dlrover_amd.common.log import logger
from dlrover_amd.common.multi_process import SharedQueue
from dlrover_amd.common.storage import (
    CheckpointStorage,
    PosixDiskStorage,
    write_tracker_step,
)
from dlrover_amd.trainer.flash_checkpoint.shm_handler import (
    SharedMemoryHandler,
    shm_segment_name,
)


def _done_file(path: str, global_rank: int) -> str:
    return os.path.join(
        path, f"{CheckpointConstant.DONE_FILE_PREFIX}{global_rank:05d}"
    )


def persist_shm_to_storage(
    handler: SharedMemoryHandler,
    event,
    storage: CheckpointStorage,
    checkpoint_dir: str,
    expected_shards: int,
) -> bool:
    """Serialize one rank's shm snapshot to storage and two-phase commit."""
    meta = handler.read_meta()
    if meta is None:
        logger.warning("shm %s holds no committed snapshot", handler.name)
        return False
    step = meta.step
    if meta.step != event.step:
        logger.warning(
            "shm %s holds step %s, event asked for %s — persisting what we have",
            handler.name,
            meta.step,
            event.step,
        )
        # event.path is named for event.step; the tracker commit below uses
        # meta.step — derive the directory from meta.step so they agree
        path = os.path.join(checkpoint_dir, str(step))
    else:
        path = event.path or os.path.join(checkpoint_dir, str(step))
    global_rank = meta.extra.get("global_rank", event.global_rank)
    state = handler.load_state_dict()
    storage.safe_makedirs(path)
    shard_name = meta.extra.get("shard_name") or f"rank_{global_rank:05d}.pt"
    shard_file = os.path.join(path, shard_name)
    tmp = shard_file + ".tmp"
    t0 = time.perf_counter()
    torch.save(state, tmp)
    os.replace(tmp, shard_file)
    storage.write(str(step), _done_file(path, global_rank))
    logger.info(
        "persisted shard rank=%s step=%s (%.1f s) -> %s",
        global_rank,
        step,
        time.perf_counter() - t0,
        shard_file,
    )
    _maybe_commit(storage, checkpoint_dir, path, step, expected_shards)
    return True


def _maybe_commit(
    storage: CheckpointStorage,
    checkpoint_dir: str,
    path: str,
    step: int,
    expected_shards: int,
):
    done = [
        f
        for f in storage.listdir(path)
        if f.startswith(CheckpointConstant.DONE_FILE_PREFIX)
    ]
    if len(done) >= expected_shards:
        write_tracker_step(storage, checkpoint_dir, step)
        storage.commit(step, True)
        logger.info("checkpoint step %s committed (%s shards)", step, len(done))


class AsyncCheckpointSaver:
    """Singleton saver in the agent process (ref: ckpt_saver.py:399)."""

    _instance: Optional["AsyncCheckpointSaver"] = None

    def __init__(
        self,
        checkpoint_dir: str,
        local_world_size: int,
        expected_shards: int,
        job_name: str = "",
        storage: Optional[CheckpointStorage] = None,
    ):
        self.checkpoint_dir = checkpoint_dir
        self.local_world_size = local_world_size
        self.expected_shards = expected_shards
        self._job = job_name or os.getenv("ELASTIC_JOB_NAME", "default")
        self.storage = storage or PosixDiskStorage()
        self._handlers: Dict[int, SharedMemoryHandler] = {
            r: SharedMemoryHandler(shm_segment_name(self._job, r), host_pin=False)
            for r in range(local_world_size)
        }
        self._event_queue = SharedQueue("flash_ckpt_events")
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._persisted_steps: Dict[int, int] = {}

    # -- lifecycle -------------------------------------------------------------

    @classmethod
    def start_async_saving_ckpt(cls, *args, **kwargs) -> "AsyncCheckpointSaver":
        if cls._instance is None:
            cls._instance = cls(*args, **kwargs)
            cls._instance.start()
        return cls._instance

    @classmethod
    def get_ckpt_saver(cls) -> Optional["AsyncCheckpointSaver"]:
        return cls._instance

    @classmethod
    def reset(cls):
        if cls._instance is not None:
            cls._instance.stop()
            cls._instance = None

    def start(self):
        t = threading.Thread(target=self._event_loop, name="ckpt-saver", daemon=True)
        t.start()
        self._threads.append(t)
        logger.info(
            "AsyncCheckpointSaver started: dir=%s local=%s expected=%s",
            self.checkpoint_dir,
            self.local_world_size,
            self.expected_shards,
        )

    def stop(self):
        self._stop.set()
        for t in self._threads:
            t.join(timeout=5)
        for h in self._handlers.values():
            h.close()

    def register_signal_handlers(self):
        """Persist the last shm snapshot before dying (ref: ckpt_saver.py:569)."""

        def _on_term(signum, frame):
            logger.info("signal %s: persisting shm checkpoints before exit", signum)
            try:
                self.save_shm_to_storage()
            finally:
                signal.signal(signum, signal.SIG_DFL)
                os.kill(os.getpid(), signum)

        signal.signal(signal.SIGTERM, _on_term)

    # -- event loop --------------------------------------------------------------

    def _event_loop(self):
        while not self._stop.is_set():
            try:
                event = self._event_queue.get(timeout=1.0)
            except pyqueue.Empty:
                continue
            except Exception:  # noqa: BLE001 — IPC server shutting down
                if self._stop.is_set():
                    return
                time.sleep(0.5)
                continue
            if getattr(event, "type", "") == "EXIT":
                return
            try:
                self._persist_one(event)
            except Exception:  # noqa: BLE001
                logger.exception("persist failed for step %s", getattr(event, "step", "?"))

    def _persist_one(self, event):
        handler = self._handlers.get(event.local_rank)
        if handler is None:
            logger.warning("no shm handler for local rank %s", event.local_rank)
            return
        ok = persist_shm_to_storage(
            handler, event, self.storage, self.checkpoint_dir,
            getattr(event, "expected_shards", 0) or self.expected_shards,
        )
        if ok:
            self._persisted_steps[event.local_rank] = event.step

    # -- failure-path persist ------------------------------------------------------

    def save_shm_to_storage(self):
        """Persist every local rank's committed shm snapshot (called by the
        agent on worker FAILED / SIGTERM; ref: training.py:1533)."""
        from dlrover_amd.trainer.flash_checkpoint.engine import CheckpointEvent

        for local_rank, handler in self._handlers.items():
            meta = handler.read_meta()
            if meta is None:
                continue
            if self._persisted_steps.get(local_rank) == meta.step:
                continue  # already on disk
            event = CheckpointEvent(
                step=meta.step,
                path=meta.extra.get("path", "")
                or os.path.join(self.checkpoint_dir, str(meta.step)),
                local_rank=local_rank,
                global_rank=meta.extra.get("global_rank", local_rank),
            )
            try:
                persist_shm_to_storage(
                    handler, event, self.storage, self.checkpoint_dir,
                    meta.extra.get("expected_shards") or self.expected_shards,
                )
                self._persisted_steps[local_rank] = meta.step
            except Exception:  # noqa: BLE001
                logger.exception("failure-path persist of rank %s failed", local_rank)
