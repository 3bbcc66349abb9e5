"""Master-side KV store backing MasterKVStore (the torch Store replacement
used for RCCL process-group bootstrap). Ref: master/elastic_training/
kv_store_service.py:1-45."""

import threading
from typing import Dict, List


class KVStoreService:
    def __init__(self):
        self._store: Dict[str, bytes] = {}
        self._lock = threading.Lock()
        self._cv = threading.Condition(self._lock)

    def get(self, key: str) -> bytes:
        with self._lock:
            return self._store.get(key, b"")

    def set(self, key: str, value: bytes):
        with self._cv:
            self._store[key] = value
            self._cv.notify_all()

    def add(self, key: str, amount: int) -> int:
        with self._cv:
            cur = int(self._store.get(key, b"0") or b"0")
            cur += amount
            self._store[key] = str(cur).encode()
            self._cv.notify_all()
            return cur

    def delete(self, key: str):
        with self._lock:
            self._store.pop(key, None)

    def multi_get(self, keys: List[str]) -> Dict[str, bytes]:
        with self._lock:
            return {k: self._store.get(k, b"") for k in keys}

    def multi_set(self, kvs: Dict[str, bytes]):
        with self._cv:
            self._store.update(kvs)
            self._cv.notify_all()

    def clear(self):
        with self._lock:
            self._store.clear()
