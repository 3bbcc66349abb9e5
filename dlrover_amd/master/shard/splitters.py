"""Dataset splitter variants (ref: dlrover/python/master/shard/
dataset_splitter.py — TableDatasetSplitter :146, TextDatasetSplitter :259,
StreamingDatasetSplitter :361). Same capability set, own structure:

- TableDatasetSplitter: index-range shards over a record table; when the
  full epoch would exceed ``max_shard_count`` shards in memory, shards are
  created one SUB-EPOCH at a time (bounded memory on huge tables).
- TextDatasetSplitter: per-SAMPLE index shards (a shard carries the line
  numbers it covers, optionally globally shuffled) for line-addressable
  text files.
- StreamingDatasetSplitter: unbounded source (message queue); the known
  size shrinks as shards are fetched and can be extended while running;
  checkpointable including partition offsets.
"""

import random
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from dlrover_amd.common.log import logger
from dlrover_amd.master.shard.task_manager import DatasetSplitter, Shard


@dataclass
class IndexShard(Shard):
    """A shard that names its exact sample indices (text datasets)."""

    indices: List[int] = field(default_factory=list)


class TableDatasetSplitter(DatasetSplitter):
    STORAGE_TYPE = "table"

    def __init__(
        self,
        dataset_name: str,
        dataset_size: int,
        shard_size: int,
        num_epochs: int = 1,
        shuffle: bool = False,
        max_shard_count: int = 50000,
    ):
        super().__init__(dataset_name, dataset_size, shard_size, num_epochs,
                         shuffle)
        self.max_shard_count = max_shard_count
        total = (dataset_size + shard_size - 1) // shard_size
        # sub-epochs bound the in-memory shard list for huge tables
        self._subepochs = max(1, -(-total // max_shard_count))
        self._subepoch = 0
        self._sub_records = -(-dataset_size // self._subepochs)

    def create_shards(self) -> List[Shard]:
        base = self._subepoch * self._sub_records
        end_total = min(base + self._sub_records, self.dataset_size)
        shards = []
        i = 0
        start = base
        while start < end_total:
            end = min(start + self.shard_size, end_total)
            shards.append(
                Shard(
                    f"{self.dataset_name}-e{self.epoch}.{self._subepoch}-s{i}",
                    start,
                    end,
                )
            )
            start = end
            i += 1
        if self.shuffle:
            random.Random(self.epoch * 1000 + self._subepoch).shuffle(shards)
        self._subepoch += 1
        if self._subepoch >= self._subepochs:
            self._subepoch = 0
            self._advance_epoch = True
        else:
            self._advance_epoch = False
        logger.info(
            "table splitter %s: %s shards (subepoch %s/%s)",
            self.dataset_name, len(shards), self._subepoch, self._subepochs,
        )
        return shards

    def epoch_complete_after_refill(self) -> bool:
        """True when the LAST create_shards() finished an epoch (the task
        manager only bumps the epoch then)."""
        return getattr(self, "_advance_epoch", True)


class TextDatasetSplitter(DatasetSplitter):
    STORAGE_TYPE = "text"

    def __init__(
        self,
        dataset_name: str,
        dataset_size: int,
        shard_size: int,
        num_epochs: int = 1,
        shuffle: bool = False,
    ):
        super().__init__(dataset_name, dataset_size, shard_size, num_epochs,
                         shuffle)

    def create_shards(self) -> List[Shard]:
        order = list(range(self.dataset_size))
        if self.shuffle:
            random.Random(self.epoch).shuffle(order)
        shards: List[Shard] = []
        for i, start in enumerate(range(0, self.dataset_size, self.shard_size)):
            end = min(start + self.shard_size, self.dataset_size)
            shards.append(
                IndexShard(
                    f"{self.dataset_name}-e{self.epoch}-s{i}",
                    start,
                    end,
                    indices=order[start:end],
                )
            )
        return shards


@dataclass
class PartitionOffsets:
    """Consumption offsets per stream partition (ref: PartitionOffsets)."""

    offsets: Dict[str, int] = field(default_factory=dict)

    def to_dict(self) -> dict:
        return dict(self.offsets)

    @staticmethod
    def from_dict(d: dict) -> "PartitionOffsets":
        return PartitionOffsets(dict(d or {}))


class StreamingDatasetSplitter(DatasetSplitter):
    """Unbounded source: ``dataset_size`` is the UNFETCHED record count
    (-1 = infinite); each create_shards() fetches up to ``fetch_size``
    records into shards and advances the global offset."""

    STORAGE_TYPE = "stream"

    def __init__(
        self,
        dataset_name: str,
        shard_size: int,
        partition_offset: Optional[PartitionOffsets] = None,
        dataset_size: int = -1,
        fetch_size: int = 10000,
        shuffle: bool = False,
    ):
        super().__init__(dataset_name, max(dataset_size, 0) or dataset_size,
                         shard_size, num_epochs=1, shuffle=shuffle)
        self.dataset_size = dataset_size  # -1 = infinite
        self.partition_offset = partition_offset or PartitionOffsets()
        self.fetch_size = fetch_size
        self._offset = 0

    def epoch_finished(self) -> bool:
        return self.dataset_size == 0

    def extend(self, records: int):
        """New records arrived in the stream."""
        if self.dataset_size < 0:
            return
        self.dataset_size += records

    def create_shards(self) -> List[Shard]:
        if self.dataset_size == 0:
            return []
        fetch = (
            self.fetch_size
            if self.dataset_size < 0
            else min(self.fetch_size, self.dataset_size)
        )
        shards = []
        i = 0
        start = self._offset
        end_total = self._offset + fetch
        while start < end_total:
            end = min(start + self.shard_size, end_total)
            shards.append(
                Shard(f"{self.dataset_name}-o{start}", start, end)
            )
            start = end
            i += 1
        self._offset = end_total
        if self.dataset_size > 0:
            self.dataset_size -= fetch
        self.epoch = 0  # streaming never advances epochs
        return shards

    # -- checkpoint (ref: to_checkpoint/from_checkpoint) ---------------------

    def to_checkpoint(self) -> dict:
        return {
            "dataset_name": self.dataset_name,
            "shard_size": self.shard_size,
            "dataset_size": self.dataset_size,
            "fetch_size": self.fetch_size,
            "offset": self._offset,
            "partition_offset": self.partition_offset.to_dict(),
        }

    @staticmethod
    def from_checkpoint(ckpt: dict) -> "StreamingDatasetSplitter":
        sp = StreamingDatasetSplitter(
            ckpt["dataset_name"],
            ckpt["shard_size"],
            PartitionOffsets.from_dict(ckpt.get("partition_offset")),
            dataset_size=ckpt.get("dataset_size", -1),
            fetch_size=ckpt.get("fetch_size", 10000),
        )
        sp._offset = ckpt.get("offset", 0)
        return sp


def new_dataset_splitter(
    storage_type: str,
    dataset_name: str,
    dataset_size: int,
    shard_size: int,
    num_epochs: int = 1,
    shuffle: bool = False,
) -> DatasetSplitter:
    """Factory mirroring the reference's new_dataset_splitter."""
    if storage_type == TableDatasetSplitter.STORAGE_TYPE:
        return TableDatasetSplitter(
            dataset_name, dataset_size, shard_size, num_epochs, shuffle
        )
    if storage_type == TextDatasetSplitter.STORAGE_TYPE:
        return TextDatasetSplitter(
            dataset_name, dataset_size, shard_size, num_epochs, shuffle
        )
    if storage_type == StreamingDatasetSplitter.STORAGE_TYPE:
        return StreamingDatasetSplitter(
            dataset_name, shard_size, dataset_size=dataset_size,
            shuffle=shuffle,
        )
    # default: plain index splitter
    return DatasetSplitter(
        dataset_name, dataset_size, shard_size, num_epochs, shuffle
    )
