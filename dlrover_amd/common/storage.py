"""Checkpoint storage abstraction (ref: dlrover/python/common/storage.py:24-334).

PosixDiskStorage + retention strategies. The write path is used by the
agent-side AsyncCheckpointSaver; the commit protocol (done files + tracker
file) lives in dlrover_amd.agent.ckpt_saver — this module only moves bytes.
"""

import os
import shutil
import tempfile
import threading
from abc import ABC, abstractmethod
from typing import List, Optional

from dlrover_amd.common.constants import CheckpointConstant
from dlrover_amd.common.log import logger


class CheckpointDeletionStrategy(ABC):
    @abstractmethod
    def clean_up(self, step_dirs: List[str], delete_fn) -> None:
        ...


class KeepLatestStepStrategy(CheckpointDeletionStrategy):
    """Keep the newest ``max_to_keep`` step directories (ref: storage.py:237)."""

    def __init__(self, max_to_keep: int = 3):
        self.max_to_keep = max(1, max_to_keep)

    def clean_up(self, step_dirs: List[str], delete_fn) -> None:
        def _step(d):
            try:
                return int(os.path.basename(d.rstrip("/")))
            except ValueError:
                return -1

        steps = sorted((d for d in step_dirs if _step(d) >= 0), key=_step)
        for d in steps[: -self.max_to_keep]:
            delete_fn(d)


class KeepStepIntervalStrategy(CheckpointDeletionStrategy):
    """Keep only steps that are a multiple of ``keep_interval`` (ref: storage.py:209)."""

    def __init__(self, keep_interval: int):
        self.keep_interval = max(1, keep_interval)

    def clean_up(self, step_dirs: List[str], delete_fn) -> None:
        for d in step_dirs:
            try:
                step = int(os.path.basename(d.rstrip("/")))
            except ValueError:
                continue
            if step % self.keep_interval != 0:
                delete_fn(d)


class CheckpointStorage(ABC):
    @abstractmethod
    def write(self, content, path: str) -> None:
        ...

    @abstractmethod
    def read(self, path: str, mode: str = "rb"):
        ...

    @abstractmethod
    def safe_rmtree(self, path: str) -> None:
        ...

    @abstractmethod
    def safe_makedirs(self, path: str) -> None:
        ...

    @abstractmethod
    def exists(self, path: str) -> bool:
        ...

    @abstractmethod
    def listdir(self, path: str) -> List[str]:
        ...

    def commit(self, step: int, success: bool) -> None:  # noqa: B027 — optional hook
        pass


class PosixDiskStorage(CheckpointStorage):
    """Local / network-posix filesystem storage (ref: storage.py:128)."""

    def write(self, content, path: str) -> None:
        self.safe_makedirs(os.path.dirname(path))
        mode = "wb" if isinstance(content, (bytes, bytearray, memoryview)) else "w"
        # write-then-rename so a crashed writer never leaves a torn file
        d = os.path.dirname(path) or "."
        fd, tmp = tempfile.mkstemp(dir=d, prefix=".tmp_dlrover_")
        try:
            with os.fdopen(fd, mode) as f:
                f.write(content)
                f.flush()
                os.fsync(f.fileno())
            os.replace(tmp, path)
        except BaseException:
            try:
                os.unlink(tmp)
            except OSError:
                pass
            raise

    def write_stream(self, path: str):
        """Open a raw stream for large sequential writes (shm → disk)."""
        self.safe_makedirs(os.path.dirname(path))
        return open(path, "wb")

    def read(self, path: str, mode: str = "rb"):
        if not os.path.exists(path):
            return None
        with open(path, mode) as f:
            return f.read()

    def safe_rmtree(self, path: str) -> None:
        try:
            if os.path.isdir(path):
                shutil.rmtree(path, ignore_errors=True)
            elif os.path.exists(path):
                os.unlink(path)
        except OSError as e:
            logger.warning("rmtree(%s) failed: %s", path, e)

    def safe_makedirs(self, path: str) -> None:
        if path:
            os.makedirs(path, exist_ok=True)

    def exists(self, path: str) -> bool:
        return os.path.exists(path)

    def listdir(self, path: str) -> List[str]:
        try:
            return sorted(os.listdir(path))
        except FileNotFoundError:
            return []


class PosixStorageWithDeletion(PosixDiskStorage):
    """Applies a retention strategy after each commit (ref: storage.py:264)."""

    def __init__(self, checkpoint_dir: str, strategy: CheckpointDeletionStrategy):
        super().__init__()
        self.checkpoint_dir = checkpoint_dir
        self.strategy = strategy
        self._lock = threading.Lock()

    def commit(self, step: int, success: bool) -> None:
        if not success:
            return
        with self._lock:
            dirs = [
                os.path.join(self.checkpoint_dir, d)
                for d in self.listdir(self.checkpoint_dir)
                if d.isdigit()
            ]
            self.strategy.clean_up(dirs, self.safe_rmtree)


def get_checkpoint_storage(
    checkpoint_dir: str = "",
    deletion_strategy: Optional[CheckpointDeletionStrategy] = None,
) -> CheckpointStorage:
    """Factory (ref: storage.py:326)."""
    if deletion_strategy and checkpoint_dir:
        return PosixStorageWithDeletion(checkpoint_dir, deletion_strategy)
    return PosixDiskStorage()


def read_tracker_step(checkpoint_dir: str) -> int:
    """Read the commit point: the step recorded in dlrover_latest.txt."""
    p = os.path.join(checkpoint_dir, CheckpointConstant.TRACKER_FILE)
    try:
        with open(p) as f:
            return int(f.read().strip())
    except (FileNotFoundError, ValueError):
        return -1


def write_tracker_step(storage: CheckpointStorage, checkpoint_dir: str, step: int):
    storage.write(str(step), os.path.join(checkpoint_dir, CheckpointConstant.TRACKER_FILE))
