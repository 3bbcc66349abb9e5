"""Worker-side dynamic-sharding client (ref: dlrover/python/elastic_agent/
sharding/client.py:29-331 — ShardingClient + IndexShardingClient).

Workers pull sample-index ranges (shards) from the master's TaskManager and
report completion; a dead worker's in-flight shards are re-queued by the
master, so elasticity never drops data.
"""

import queue
import threading
from typing import Optional

from dlrover_amd.agent.master_client import MasterClient
from dlrover_amd.common import comm


class ShardingClient:
    def __init__(
        self,
        dataset_name: str,
        dataset_size: int,
        batch_size: int,
        num_epochs: int = 1,
        shard_size: int = 0,
        shuffle: bool = False,
        client: Optional[MasterClient] = None,
    ):
        self.dataset_name = dataset_name
        self._client = client or MasterClient.singleton_instance()
        self._client.report_dataset_params(
            comm.DatasetShardParams(
                dataset_name=dataset_name,
                dataset_size=dataset_size,
                shard_size=shard_size or batch_size,
                batch_size=batch_size,
                num_epochs=num_epochs,
                shuffle=shuffle,
            )
        )
        self._current: Optional[comm.Task] = None
        self._lock = threading.Lock()

    def fetch_shard(self) -> Optional[comm.Task]:
        task = self._client.get_task(self.dataset_name)
        if task.empty:
            return None
        with self._lock:
            self._current = task
        return task

    def report_batch_done(self, task: Optional[comm.Task] = None, success: bool = True):
        with self._lock:
            task = task or self._current
            self._current = None
        if task is not None and not task.empty:
            self._client.report_task_result(
                self.dataset_name, task.task_id, success=success
            )

    def checkpoint(self) -> str:
        return self._client.get_shard_checkpoint(self.dataset_name)

    def restore(self, content: str):
        self._client.report_shard_checkpoint(self.dataset_name, content)


class IndexShardingClient(ShardingClient):
    """Streams per-sample indices out of fetched shards — plugs under a
    torch Dataset/Sampler (ref: client.py:232)."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._indices: "queue.Queue[int]" = queue.Queue()
        self._shard_of_index: dict = {}

    def fetch_sample_index(self) -> Optional[int]:
        if self._indices.empty():
            task = self.fetch_shard()
            if task is None:
                return None
            for i in range(task.start, task.end):
                self._indices.put(i)
                self._shard_of_index[i] = task
        try:
            return self._indices.get_nowait()
        except queue.Empty:
            return None

    def report_sample_done(self, index: int):
        task = self._shard_of_index.pop(index, None)
        if task is not None and all(
            i not in self._shard_of_index for i in range(task.start, task.end)
        ):
            self.report_batch_done(task)
