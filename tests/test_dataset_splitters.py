"""Splitter variants (ref test model: dlrover test_dataset_splitter.py)."""

from dlrover_amd.master.shard.splitters import (
    IndexShard,
    PartitionOffsets,
    StreamingDatasetSplitter,
    TableDatasetSplitter,
    TextDatasetSplitter,
    new_dataset_splitter,
)
from dlrover_amd.master.shard.task_manager import DatasetManager


def test_table_splitter_basic_ranges():
    sp = TableDatasetSplitter("t", dataset_size=10000, shard_size=100,
                              num_epochs=1)
    shards = sp.create_shards()
    assert len(shards) == 100
    assert shards[0].start == 0 and shards[0].end == 100
    assert shards[-1].end == 10000
    # full coverage, no overlap
    covered = sorted((s.start, s.end) for s in shards)
    assert all(a[1] == b[0] for a, b in zip(covered, covered[1:]))


def test_table_splitter_subepochs_bound_memory():
    sp = TableDatasetSplitter("t", dataset_size=1000, shard_size=10,
                              num_epochs=1, max_shard_count=25)
    seen = []
    rounds = 0
    while True:
        shards = sp.create_shards()
        seen.extend(shards)
        rounds += 1
        assert len(shards) <= 25
        if sp.epoch_complete_after_refill():
            break
    assert rounds == 4  # 100 shards / 25 per subepoch
    assert len(seen) == 100
    assert {(s.start, s.end) for s in seen} == {
        (i * 10, i * 10 + 10) for i in range(100)
    }


def test_table_splitter_shuffle_changes_order():
    a = TableDatasetSplitter("t", 1000, 10, shuffle=True)
    b = TableDatasetSplitter("t", 1000, 10, shuffle=False)
    sa = a.create_shards()
    sb = b.create_shards()
    assert [s.start for s in sa] != [s.start for s in sb]
    assert {s.start for s in sa} == {s.start for s in sb}


def test_text_splitter_indices_cover_all():
    sp = TextDatasetSplitter("f.txt", dataset_size=95, shard_size=10,
                             shuffle=True)
    shards = sp.create_shards()
    assert len(shards) == 10
    assert all(isinstance(s, IndexShard) for s in shards)
    allidx = [i for s in shards for i in s.indices]
    assert sorted(allidx) == list(range(95))
    assert len(shards[-1].indices) == 5


def test_streaming_splitter_consumes_and_extends():
    sp = StreamingDatasetSplitter("q", shard_size=8,
                                  partition_offset=PartitionOffsets(),
                                  dataset_size=20, fetch_size=16)
    s1 = sp.create_shards()
    assert [(s.start, s.end) for s in s1] == [(0, 8), (8, 16)]
    assert sp.dataset_size == 4
    s2 = sp.create_shards()
    assert [(s.start, s.end) for s in s2] == [(16, 20)]
    assert sp.epoch_finished()
    sp.extend(10)
    assert not sp.epoch_finished()
    s3 = sp.create_shards()
    assert s3[0].start == 20


def test_streaming_splitter_checkpoint_roundtrip():
    sp = StreamingDatasetSplitter("q", shard_size=5, dataset_size=100,
                                  fetch_size=10)
    sp.partition_offset.offsets["p0"] = 42
    sp.create_shards()
    ckpt = sp.to_checkpoint()
    sp2 = StreamingDatasetSplitter.from_checkpoint(ckpt)
    assert sp2.dataset_size == 90
    assert sp2.partition_offset.offsets == {"p0": 42}
    nxt = sp2.create_shards()
    assert nxt[0].start == 10  # resumes at the persisted offset


def test_manager_with_table_subepochs():
    sp = TableDatasetSplitter("t", dataset_size=100, shard_size=10,
                              num_epochs=2, max_shard_count=5)
    mgr = DatasetManager(sp)
    got = []
    while True:
        t = mgr.get_task(node_id=0)
        if t is None:
            break
        got.append(t)
        mgr.report_result(t.task_id, success=True)
    # 10 shards/epoch x 2 epochs, delivered in bounded subepoch batches
    assert len(got) == 20
    epochs = {t.epoch for t in got}
    assert epochs == {0, 1}


def test_factory():
    assert isinstance(new_dataset_splitter("table", "d", 10, 2),
                      TableDatasetSplitter)
    assert isinstance(new_dataset_splitter("text", "d", 10, 2),
                      TextDatasetSplitter)
    assert isinstance(new_dataset_splitter("stream", "d", 10, 2),
                      StreamingDatasetSplitter)
