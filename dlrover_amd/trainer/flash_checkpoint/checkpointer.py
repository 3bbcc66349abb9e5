"""User-facing checkpointer base (ref: dlrover/trainer/torch/flash_checkpoint/
checkpointer.py — StorageType, Checkpointer ABC)."""

from abc import ABC, abstractmethod
from typing import Optional


class StorageType:
    MEMORY = 0
    DISK = 1


class Checkpointer(ABC):
    """save_checkpoint(step, state_dict, path, storage_type) /
    load_checkpoint(resume_path) — the two-call API the reference exposes."""

    @abstractmethod
    def save_checkpoint(
        self,
        step: int,
        state_dict: Optional[dict] = None,
        path: str = "",
        storage_type: int = StorageType.DISK,
    ) -> float:
        """Returns blocking seconds (time training was stalled)."""

    @abstractmethod
    def load_checkpoint(self, resume_path: str = "") -> Optional[dict]:
        ...

    def wait_latest_checkpoint(self, timeout: int = 600):  # noqa: B027
        """Block until async persistence of the newest snapshot completes."""

    def close(self):  # noqa: B027
        pass
