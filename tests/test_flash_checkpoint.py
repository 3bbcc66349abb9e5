"""Flash checkpoint: shm snapshot/restore, commit protocol, standalone
persistence, DDP checkpointer round-trip — all CPU."""

import os
import time

import pytest
import torch

from dlrover_amd.common.storage import read_tracker_step
from dlrover_amd.trainer.flash_checkpoint import DdpCheckpointer, StorageType
from dlrover_amd.trainer.flash_checkpoint.shm_handler import (
    SharedMemoryHandler,
    plan_layout,
    traverse_state_dict,
)


@pytest.fixture()
def shm_name(monkeypatch):
    name = f"dlrover_test_{os.getpid()}_{time.time_ns()}"
    yield name


def _demo_state():
    return {
        "step": 7,
        "model": {
            "w": torch.randn(4, 8),
            "b": torch.randn(8, dtype=torch.float64),
            "emb": torch.randn(3, 2).bfloat16(),
        },
        "optimizer": {
            "state": {0: {"exp_avg": torch.randn(4, 8), "step": 12}},
            "param_groups": [{"lr": 0.1, "params": [0]}],
        },
    }


def test_traverse_and_layout():
    sd = _demo_state()
    leaves = dict(traverse_state_dict(sd))
    assert ("model", "w") in leaves
    assert ("optimizer", "state", 0, "step") in leaves
    meta = plan_layout(sd)
    assert len(meta.tensors) == 4
    assert meta.payload_bytes % 64 == 0
    # offsets are 64-aligned and non-overlapping
    offs = sorted((t.offset, t.nbytes) for t in meta.tensors)
    for (o1, n1), (o2, _) in zip(offs, offs[1:]):
        assert o1 + n1 <= o2 and o2 % 64 == 0


def test_shm_save_load_roundtrip(shm_name):
    h = SharedMemoryHandler(shm_name)
    try:
        sd = _demo_state()
        blocking = h.save_state_dict(7, sd, extra={"path": "/tmp/x"})
        assert blocking >= 0
        assert h.committed_step() == 7
        out = h.load_state_dict()
        assert out["step"] == 7
        torch.testing.assert_close(out["model"]["w"], sd["model"]["w"])
        torch.testing.assert_close(out["model"]["b"], sd["model"]["b"])
        torch.testing.assert_close(
            out["model"]["emb"].float(), sd["model"]["emb"].float()
        )
        assert out["optimizer"]["param_groups"][0]["lr"] == 0.1
        assert out["optimizer"]["state"][0]["step"] == 12
        meta = h.read_meta()
        assert meta.extra["path"] == "/tmp/x"
    finally:
        h.unlink()


def test_shm_overwrite_and_grow(shm_name):
    h = SharedMemoryHandler(shm_name)
    try:
        h.save_state_dict(1, {"t": torch.zeros(10)})
        h.save_state_dict(2, {"t": torch.ones(100000)})  # forces resize
        out = h.load_state_dict()
        assert h.committed_step() == 2
        assert out["t"].numel() == 100000
    finally:
        h.unlink()


def test_empty_segment_reads_none(shm_name):
    h = SharedMemoryHandler(shm_name)
    assert h.load_state_dict() is None
    assert h.committed_step() == 0


def test_ddp_checkpointer_roundtrip(tmp_path, monkeypatch):
    monkeypatch.setenv("ELASTIC_JOB_NAME", f"t{time.time_ns()}")
    torch.manual_seed(0)
    model = torch.nn.Linear(8, 8)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    ckdir = str(tmp_path / "ckpt")
    cp = DdpCheckpointer(ckdir, model, opt)
    try:
        # train a step so optimizer has state
        loss = model(torch.randn(4, 8)).pow(2).mean()
        loss.backward()
        opt.step()
        blocking = cp.save_checkpoint(10, storage_type=StorageType.DISK)
        assert blocking >= 0
        cp.wait_latest_checkpoint()
        # two-phase commit artifacts
        assert read_tracker_step(ckdir) == 10
        assert os.path.exists(os.path.join(ckdir, "10", "rank_00000.pt"))
        assert os.path.exists(os.path.join(ckdir, "10", ".done_00000"))

        # memory-only save then restore from shm
        with torch.no_grad():
            saved_w = model.weight.clone()
            cp.save_checkpoint(20, storage_type=StorageType.MEMORY)
            model.weight.add_(1.0)
        out = cp.load_checkpoint()
        assert out["step"] == 20
        torch.testing.assert_close(model.weight, saved_w)
    finally:
        cp.close()
        cp.engine.shm_handler.unlink()


def test_restore_from_disk_after_shm_gone(tmp_path, monkeypatch):
    monkeypatch.setenv("ELASTIC_JOB_NAME", f"t{time.time_ns()}")
    model = torch.nn.Linear(4, 4)
    ckdir = str(tmp_path / "ck")
    cp = DdpCheckpointer(ckdir, model)
    try:
        with torch.no_grad():
            saved = model.weight.clone()
        cp.save_checkpoint(5, storage_type=StorageType.DISK)
        cp.wait_latest_checkpoint()
    finally:
        cp.close()
        cp.engine.shm_handler.unlink()  # simulate node loss of shm

    with torch.no_grad():
        model.weight.zero_()
    cp2 = DdpCheckpointer(ckdir, model)
    try:
        out = cp2.load_checkpoint()
        assert out is not None and out["step"] == 5
        torch.testing.assert_close(model.weight, saved)
    finally:
        cp2.close()
        cp2.engine.shm_handler.unlink()


def test_derived_param_omission_and_rederive(tmp_path):
    """bf16 params with an fp32 master are OMITTED from the snapshot and
    re-derived from the restored master (the GPU saves 16 GB on 8B; the
    logic is dtype-driven so a bf16 CPU model exercises it end-to-end)."""
    import torch

    from dlrover_amd.models import LlamaConfig, LlamaForCausalLM
    from dlrover_amd.ops import FusedAdamW
    from dlrover_amd.trainer.flash_checkpoint.engine import (
        ShardedCheckpointEngine,
    )

    torch.manual_seed(0)
    model = LlamaForCausalLM(LlamaConfig.tiny()).bfloat16()
    opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.0)
    ids = torch.randint(0, 512, (2, 16))
    model(ids, ids.clone()).backward()
    opt.step()
    opt.zero_grad()

    eng = ShardedCheckpointEngine(str(tmp_path / "ckpt"))
    sd = eng.gather_state_dict(model, opt)
    n_params = sum(1 for _ in model.named_parameters())
    assert len(sd["_derived"]) == n_params  # every param has a master
    for name in sd["_derived"]:
        assert name not in sd["model"]  # omitted from the payload
    # buffers (rope tables) still present
    assert any("rope" in k for k in sd["model"])

    before = {n: p.detach().clone() for n, p in model.named_parameters()}
    eng.save_to_memory(3, sd)
    with torch.no_grad():
        for p in model.parameters():
            p.add_(1.0)
    restored = eng.restore_into(model, opt)
    assert restored is not None
    for n, p in model.named_parameters():
        assert torch.equal(p.detach(), before[n]), n
    eng.shm_handler.unlink()


def test_derived_disabled_by_env(tmp_path, monkeypatch):
    import torch

    from dlrover_amd.models import LlamaConfig, LlamaForCausalLM
    from dlrover_amd.ops import FusedAdamW
    from dlrover_amd.trainer.flash_checkpoint.engine import (
        ShardedCheckpointEngine,
    )

    monkeypatch.setenv("DLROVER_CKPT_DERIVED", "0")
    model = LlamaForCausalLM(LlamaConfig.tiny()).bfloat16()
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    ids = torch.randint(0, 512, (1, 16))
    model(ids, ids.clone()).backward()
    opt.step()
    eng = ShardedCheckpointEngine(str(tmp_path / "ckpt"))
    sd = eng.gather_state_dict(model, opt)
    assert sd["_derived"] == {}
    assert all(n in sd["model"] for n, _ in model.named_parameters())


def test_save_to_memory_skips_when_peer_not_ready(monkeypatch):
    """A rank that cannot proceed must not strand peers writing
    inconsistent steps (ref engine.py:60 readiness all-reduce)."""
    import uuid

    import torch

    from dlrover_amd.trainer.flash_checkpoint.engine import (
        ShardedCheckpointEngine,
    )

    eng = ShardedCheckpointEngine(f"/tmp/nr{uuid.uuid4().hex[:6]}")
    # single-process: force the readiness check to report a bad peer
    monkeypatch.setattr(eng, "_check_all_ranks_ready", lambda ready: False)
    blocking = eng.save_to_memory(3, {"t": torch.ones(2)})
    assert blocking == 0.0
    assert eng.shm_handler.committed_step() == 0  # nothing written
    eng.shm_handler.unlink()


def test_save_step_mismatch_raises(monkeypatch):
    import uuid

    import pytest
    import torch

    from dlrover_amd.trainer.flash_checkpoint.engine import (
        ShardedCheckpointEngine,
    )

    eng = ShardedCheckpointEngine(f"/tmp/sm{uuid.uuid4().hex[:6]}")
    monkeypatch.setattr(eng, "_check_step_consistent", lambda step: False)
    with pytest.raises(RuntimeError, match="differs across ranks"):
        eng.save_to_memory(3, {"t": torch.ones(2)})
    eng.shm_handler.unlink()
