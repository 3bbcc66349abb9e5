"""Dynamic data sharding: the master splits the sample-index space into
shards, dispatches them as tasks to workers, and re-queues shards of dead
workers so no data is lost under elasticity.

Parity target: ref dlrover/python/master/shard/dataset_splitter.py:92-430
(TableDatasetSplitter/TextDatasetSplitter/StreamingDatasetSplitter),
batch_dataset_manager.py:29 (doing/todo bookkeeping + recovery) and
task_manager.py:35.
"""

import json
import threading
from dataclasses import dataclass
from typing import Dict, List, Optional

from dlrover_amd.common import comm
from dlrover_amd.common.log import logger


@dataclass
class Shard:
    name: str
    start: int
    end: int


@dataclass
class Task:
    task_id: int
    shard: Shard
    epoch: int
    node_id: int = -1


class DatasetSplitter:
    """Split [0, dataset_size) into shards, optionally shuffled, per epoch."""

    def __init__(
        self,
        dataset_name: str,
        dataset_size: int,
        shard_size: int,
        num_epochs: int = 1,
        shuffle: bool = False,
    ):
        if shard_size <= 0:
            raise ValueError("shard_size must be positive")
        self.dataset_name = dataset_name
        self.dataset_size = dataset_size
        self.shard_size = shard_size
        self.num_epochs = num_epochs
        self.shuffle = shuffle
        self.epoch = 0

    def create_shards(self) -> List[Shard]:
        shards = []
        for i, start in enumerate(range(0, self.dataset_size, self.shard_size)):
            end = min(start + self.shard_size, self.dataset_size)
            shards.append(Shard(f"{self.dataset_name}-e{self.epoch}-s{i}", start, end))
        if self.shuffle:
            import random

            rng = random.Random(self.epoch)
            rng.shuffle(shards)
        return shards

    def epoch_finished(self) -> bool:
        return self.epoch >= self.num_epochs


class DatasetManager:
    """todo/doing bookkeeping for one dataset (ref: batch_dataset_manager.py)."""

    def __init__(self, splitter: DatasetSplitter):
        self.splitter = splitter
        self._todo: List[Task] = []
        self._doing: Dict[int, Task] = {}
        self._task_id = 0
        self._completed = 0
        self._lock = threading.Lock()

    def _refill_locked(self):
        if self._todo or self._doing:
            return
        if self.splitter.epoch_finished():
            return
        for shard in self.splitter.create_shards():
            self._todo.append(Task(self._task_id, shard, self.splitter.epoch))
            self._task_id += 1
        # sub-epoch splitters (table variant) only finish an epoch after
        # their LAST refill; streaming never advances epochs
        done_hook = getattr(self.splitter, "epoch_complete_after_refill", None)
        if done_hook is None or done_hook():
            self.splitter.epoch += 1

    def get_task(self, node_id: int) -> Optional[Task]:
        with self._lock:
            self._refill_locked()
            if not self._todo:
                return None
            task = self._todo.pop(0)
            task.node_id = node_id
            self._doing[task.task_id] = task
            return task

    def report_result(self, task_id: int, success: bool):
        with self._lock:
            task = self._doing.pop(task_id, None)
            if task is None:
                return
            if success:
                self._completed += 1
            else:
                self._todo.insert(0, task)

    def recover_tasks(self, node_id: int):
        """Re-queue in-flight shards of a dead worker (ref:
        task_manager.recover_tasks)."""
        with self._lock:
            lost = [t for t in self._doing.values() if t.node_id == node_id]
            for t in lost:
                del self._doing[t.task_id]
                self._todo.insert(0, t)
            if lost:
                logger.info(
                    "recovered %s shards from dead node %s", len(lost), node_id
                )

    def finished(self) -> bool:
        with self._lock:
            return (
                not self._todo and not self._doing and self.splitter.epoch_finished()
            )

    # -- checkpointable shard state (ref: DatasetShardCheckpoint :60) -----------

    def checkpoint(self) -> str:
        with self._lock:
            return json.dumps(
                {
                    "epoch": self.splitter.epoch,
                    "todo": [
                        [t.task_id, t.shard.name, t.shard.start, t.shard.end, t.epoch]
                        for t in self._todo + list(self._doing.values())
                    ],
                    "task_id": self._task_id,
                    "completed": self._completed,
                }
            )

    def restore(self, content: str):
        data = json.loads(content)
        with self._lock:
            self.splitter.epoch = data["epoch"]
            self._task_id = data["task_id"]
            self._completed = data.get("completed", 0)
            self._doing.clear()
            self._todo = [
                Task(tid, Shard(name, start, end), epoch)
                for tid, name, start, end, epoch in data["todo"]
            ]


class TaskManager:
    """All datasets of a job (ref: master/shard/task_manager.py:35)."""

    def __init__(self):
        self._datasets: Dict[str, DatasetManager] = {}
        self._lock = threading.Lock()

    def new_dataset(self, params: comm.DatasetShardParams):
        with self._lock:
            if params.dataset_name in self._datasets:
                return
            splitter = DatasetSplitter(
                params.dataset_name,
                params.dataset_size,
                params.shard_size or max(params.batch_size, 1),
                num_epochs=params.num_epochs,
                shuffle=params.shuffle,
            )
            self._datasets[params.dataset_name] = DatasetManager(splitter)
            logger.info(
                "dataset %s registered: size=%s shard=%s epochs=%s",
                params.dataset_name,
                params.dataset_size,
                splitter.shard_size,
                params.num_epochs,
            )

    def get_task(self, dataset_name: str, node_id: int) -> comm.Task:
        ds = self._datasets.get(dataset_name)
        if ds is None:
            return comm.Task()
        task = ds.get_task(node_id)
        if task is None:
            return comm.Task()
        return comm.Task(
            task_id=task.task_id,
            dataset_name=dataset_name,
            shard_name=task.shard.name,
            start=task.shard.start,
            end=task.shard.end,
            epoch=task.epoch,
        )

    def report_task_result(self, result: comm.TaskResult):
        ds = self._datasets.get(result.dataset_name)
        if ds is not None:
            ds.report_result(result.task_id, result.success)

    def recover_tasks(self, node_id: int):
        with self._lock:
            for ds in self._datasets.values():
                ds.recover_tasks(node_id)

    def checkpoint_dataset(self, dataset_name: str) -> str:
        ds = self._datasets.get(dataset_name)
        return ds.checkpoint() if ds is not None else ""

    def restore_dataset(self, dataset_name: str, content: str):
        ds = self._datasets.get(dataset_name)
        if ds is not None and content:
            ds.restore(content)

    def finished(self) -> bool:
        with self._lock:
            return all(ds.finished() for ds in self._datasets.values())
