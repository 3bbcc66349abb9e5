"""Agent-mode flash checkpoint: training-process engine + agent-process
AsyncCheckpointSaver wired through the real IPC server (single process,
both roles in threads — the e2e tests cover true process separation)."""

import os
import time
import uuid

import pytest
import torch

from dlrover_amd.agent.ckpt_saver import AsyncCheckpointSaver
from dlrover_amd.common.multi_process import IPCServer, ipc_socket_path
from dlrover_amd.common.storage import read_tracker_step
from dlrover_amd.trainer.flash_checkpoint.engine import FullCheckpointEngine


@pytest.fixture()
def agent_env(tmp_path, monkeypatch):
    job = f"aflow{uuid.uuid4().hex[:6]}"
    monkeypatch.setenv("ELASTIC_JOB_NAME", job)
    monkeypatch.setenv("DLROVER_IPC_SOCKET_DIR", str(tmp_path / "ipc"))
    server = IPCServer().start()
    saver = AsyncCheckpointSaver(
        checkpoint_dir=str(tmp_path / "ckpt"),
        local_world_size=1,
        expected_shards=1,
        job_name=job,
    )
    saver.start()
    yield tmp_path, saver
    AsyncCheckpointSaver._instance = None
    saver.stop()
    server.stop()


def test_agent_mode_save_persists_and_commits(agent_env):
    tmp_path, saver = agent_env
    engine = FullCheckpointEngine(str(tmp_path / "ckpt"))
    assert engine._agent_mode, "engine must detect the agent IPC socket"
    sd = {"step": 21, "model": {"w": torch.randn(8, 8)}}
    engine.save_to_storage(21, sd)
    deadline = time.time() + 30
    while time.time() < deadline:
        if read_tracker_step(str(tmp_path / "ckpt")) == 21:
            break
        time.sleep(0.2)
    assert read_tracker_step(str(tmp_path / "ckpt")) == 21
    assert os.path.exists(tmp_path / "ckpt" / "21" / "rank_00000.pt")
    engine.close()
    engine.shm_handler.unlink()


def test_failure_path_persist(agent_env):
    """Worker 'dies' after a MEMORY-only save; the agent's failure path
    (save_shm_to_storage, as _invoke_run calls on FAILED) must persist it."""
    tmp_path, saver = agent_env
    engine = FullCheckpointEngine(str(tmp_path / "ckpt"))
    sd = {"step": 33, "model": {"w": torch.ones(4)}}
    engine.save_to_memory(33, sd)  # never asked for DISK
    # agent-side handlers attach to the same segments
    saver.save_shm_to_storage()
    assert read_tracker_step(str(tmp_path / "ckpt")) == 33
    data = torch.load(tmp_path / "ckpt" / "33" / "rank_00000.pt", weights_only=False)
    torch.testing.assert_close(data["model"]["w"], torch.ones(4))
    engine.close()
    engine.shm_handler.unlink()


def test_reader_reattaches_after_segment_regrowth(tmp_path, monkeypatch):
    """a writer that outgrows its segment unlinks+recreates it;
    a cached reader mapping must re-attach instead of silently reading the
    dead segment forever."""
    import uuid

    import torch

    from dlrover_amd.trainer.flash_checkpoint.shm_handler import (
        SharedMemoryHandler,
        shm_segment_name,
    )

    name = shm_segment_name(f"regrow{uuid.uuid4().hex[:6]}", 0)
    writer = SharedMemoryHandler(name, host_pin=False)
    writer.save_state_dict(1, {"t": torch.ones(8)})

    reader = SharedMemoryHandler(name, host_pin=False)
    assert reader.committed_step() == 1

    # writer outgrows the segment -> unlink + recreate bigger
    writer.save_state_dict(2, {"t": torch.ones(1 << 22)})
    assert reader.committed_step() == 2  # re-attached, not stale
    meta = reader.read_meta()
    assert meta is not None and meta.tensors[0].nbytes == (1 << 22) * 4
    writer.unlink()


def test_persist_mismatched_step_keeps_tracker_consistent(tmp_path):
    """when shm holds a DIFFERENT step than the event, the shard
    directory must be derived from the shm step so the tracker never points
    at a directory that does not exist."""
    import uuid

    import torch

    from dlrover_amd.agent.ckpt_saver import persist_shm_to_storage
    from dlrover_amd.common.storage import PosixDiskStorage, read_tracker_step
    from dlrover_amd.trainer.flash_checkpoint.engine import CheckpointEvent
    from dlrover_amd.trainer.flash_checkpoint.shm_handler import (
        SharedMemoryHandler,
        shm_segment_name,
    )

    name = shm_segment_name(f"mm{uuid.uuid4().hex[:6]}", 0)
    h = SharedMemoryHandler(name, host_pin=False)
    h.save_state_dict(5, {"t": torch.ones(4)},
                      extra={"global_rank": 0, "expected_shards": 1})
    ckpt_dir = str(tmp_path / "ckpt")
    event = CheckpointEvent(step=9, path=f"{ckpt_dir}/9", local_rank=0,
                            global_rank=0)
    ok = persist_shm_to_storage(h, event, PosixDiskStorage(), ckpt_dir, 1)
    assert ok
    step = read_tracker_step(ckpt_dir)
    assert step == 5
    import os

    assert os.path.exists(os.path.join(ckpt_dir, "5", "rank_00000.pt"))
    h.unlink()
