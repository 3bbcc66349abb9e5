"""MasterKVStore: a torch.distributed.Store whose get/set/add/wait go to the
job master's KV service — replacing TCPStore for process-group bootstrap so
rendezvous survives node churn (ref: dlrover/python/elastic_agent/torch/
master_kv_store.py:1-176).
"""

import time
from datetime import timedelta
from typing import List

import torch.distributed as dist

from dlrover_amd.agent.master_client import MasterClient


class MasterKVStore(dist.Store):
    def __init__(
        self,
        prefix: str,
        client: MasterClient = None,
        timeout: timedelta = timedelta(seconds=300),
    ):
        super().__init__()
        self.prefix = prefix
        self.client = client or MasterClient.singleton_instance()
        self._timeout = timeout

    def _key(self, key: str) -> str:
        return f"{self.prefix}/{key}"

    # -- Store interface ---------------------------------------------------------

    def set(self, key, value):
        if isinstance(value, str):
            value = value.encode()
        self.client.kv_store_set(self._key(key), bytes(value))

    def get(self, key) -> bytes:
        deadline = time.monotonic() + self._timeout.total_seconds()
        k = self._key(key)
        while True:
            v = self.client.kv_store_get(k)
            if v:
                return v
            if time.monotonic() > deadline:
                raise LookupError(f"MasterKVStore: key {k} not set within timeout")
            time.sleep(0.1)

    def add(self, key, num: int) -> int:
        return self.client.kv_store_add(self._key(key), num)

    def check(self, keys: List[str]) -> bool:
        vals = self.client.kv_store_multi_get([self._key(k) for k in keys])
        return all(v for v in vals.values())

    def wait(self, keys: List[str], override_timeout: timedelta = None):
        timeout = (override_timeout or self._timeout).total_seconds()
        deadline = time.monotonic() + timeout
        while not self.check(keys):
            if time.monotonic() > deadline:
                raise LookupError(f"MasterKVStore: keys {keys} not all set in time")
            time.sleep(0.1)

    def delete_key(self, key) -> bool:
        self.client.kv_store_delete(self._key(key))
        return True

    def set_timeout(self, timeout: timedelta):
        self._timeout = timeout

    def compare_set(self, key, expected, desired) -> bytes:
        # optimistic: used rarely (torch barrier impls); emulate via get+set
        cur = self.client.kv_store_get(self._key(key))
        exp = expected.encode() if isinstance(expected, str) else bytes(expected)
        des = desired.encode() if isinstance(desired, str) else bytes(desired)
        if cur == exp or (not cur and not exp):
            self.client.kv_store_set(self._key(key), des)
            return des
        return cur

    def num_keys(self) -> int:  # pragma: no cover - informational only
        return 0
