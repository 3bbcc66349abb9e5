"""Named sync groups / barriers across workers (ref: master/elastic_training/
sync_service.py:1-117)."""

import threading
from typing import Dict, Set


class SyncService:
    def __init__(self):
        self._lock = threading.Lock()
        self._joined: Dict[str, Set[int]] = {}
        self._finished: Set[str] = set()
        self._barriers: Set[str] = set()

    def join_sync(self, sync_name: str, node_id: int):
        with self._lock:
            self._joined.setdefault(sync_name, set()).add(node_id)

    def sync_finished(self, sync_name: str):
        with self._lock:
            self._finished.add(sync_name)

    def is_sync_finished(self, sync_name: str) -> bool:
        with self._lock:
            return sync_name in self._finished

    def joined_count(self, sync_name: str) -> int:
        with self._lock:
            return len(self._joined.get(sync_name, ()))

    def notify_barrier(self, barrier_name: str):
        with self._lock:
            self._barriers.add(barrier_name)

    def barrier_reached(self, barrier_name: str) -> bool:
        with self._lock:
            return barrier_name in self._barriers
