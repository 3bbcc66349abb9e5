"""Substrate tests: serialize, node model, storage retention/commit."""

import os
import pickle

import pytest

from dlrover_amd.common import comm
from dlrover_amd.common.constants import NodeExitReason, NodeStatus, NodeType
from dlrover_amd.common.node import Node, NodeResource
from dlrover_amd.common.serialize import dumps, loads
from dlrover_amd.common.storage import (
    KeepLatestStepStrategy,
    KeepStepIntervalStrategy,
    PosixDiskStorage,
    PosixStorageWithDeletion,
    read_tracker_step,
    write_tracker_step,
)


def test_serialize_roundtrip():
    msg = comm.JoinRendezvousRequest(node_id=3, node_rank=1, local_world_size=8)
    out = loads(dumps(msg))
    assert out == msg


def test_serialize_rejects_foreign_class():
    class Evil:
        def __reduce__(self):
            return (os.system, ("true",))

    data = pickle.dumps(Evil())
    with pytest.raises(pickle.UnpicklingError):
        loads(data)


def test_node_lifecycle():
    n = Node(NodeType.WORKER, 0, config_resource=NodeResource(gpu_num=8))
    assert n.is_alive()
    n.update_status(NodeStatus.RUNNING)
    assert n.start_time is not None
    n.update_status(NodeStatus.FAILED)
    assert not n.is_alive()
    assert n.should_relaunch()
    n.exit_reason = NodeExitReason.FATAL_ERROR
    assert not n.should_relaunch()
    assert n.is_unrecoverable_failure()


def test_node_relaunch_budget():
    n = Node(NodeType.WORKER, 0, max_relaunch_count=2)
    n.inc_relaunch_count()
    assert n.should_relaunch()
    n.inc_relaunch_count()
    assert not n.should_relaunch()
    repl = n.new_incarnation(5)
    assert repl.id == 5 and repl.rank_index == n.rank_index
    assert repl.relaunch_count == 3


def test_posix_storage_roundtrip(tmp_path):
    st = PosixDiskStorage()
    p = str(tmp_path / "a" / "b.bin")
    st.write(b"hello", p)
    assert st.read(p) == b"hello"
    assert st.read(str(tmp_path / "missing")) is None
    st.safe_rmtree(str(tmp_path / "a"))
    assert not st.exists(p)


def test_tracker_file(tmp_path):
    st = PosixDiskStorage()
    assert read_tracker_step(str(tmp_path)) == -1
    write_tracker_step(st, str(tmp_path), 50)
    assert read_tracker_step(str(tmp_path)) == 50


def test_keep_latest_strategy(tmp_path):
    for s in (10, 20, 30, 40):
        os.makedirs(tmp_path / str(s))
    st = PosixStorageWithDeletion(str(tmp_path), KeepLatestStepStrategy(max_to_keep=2))
    st.commit(40, success=True)
    assert sorted(os.listdir(tmp_path)) == ["30", "40"]


def test_keep_interval_strategy(tmp_path):
    for s in (10, 15, 20, 25):
        os.makedirs(tmp_path / str(s))
    st = PosixStorageWithDeletion(str(tmp_path), KeepStepIntervalStrategy(keep_interval=10))
    st.commit(25, success=True)
    assert sorted(os.listdir(tmp_path)) == ["10", "20"]


def test_serialize_rejects_dangerous_builtins():
    """A REDUCE of builtins.eval must NOT resolve (RCE from the wire);
    safe builtin containers still roundtrip."""
    import pickle

    import pytest

    evil = b"cbuiltins\neval\n(V1+1\ntR."
    with pytest.raises(pickle.UnpicklingError):
        loads(evil)
    assert loads(dumps({"a", 1})) == {"a", 1}
    assert loads(dumps(bytearray(b"x"))) == bytearray(b"x")


def test_keep_step_interval_strategy(tmp_path):
    """Checkpoints off the keep-interval grid are deleted once committed
    (ref storage.py KeepStepIntervalStrategy)."""
    import os

    from dlrover_amd.common.storage import (
        KeepStepIntervalStrategy,
        PosixStorageWithDeletion,
    )

    st = PosixStorageWithDeletion(
        str(tmp_path), KeepStepIntervalStrategy(keep_interval=10)
    )
    for step in (5, 10, 15, 20):
        os.makedirs(tmp_path / str(step), exist_ok=True)
        (tmp_path / str(step) / "x.pt").write_bytes(b"d")
        st.commit(step, True)
    remaining = {d.name for d in tmp_path.iterdir() if d.is_dir()}
    assert remaining == {"10", "20"}, remaining
