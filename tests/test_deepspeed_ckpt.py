"""DeepSpeedCheckpointer exercised with a MOCK deepspeed engine (the
reference's deepspeed_ckpt_test.py does the same — the library itself is
not in the image): save-to-memory / disk persist / restore roundtrip of
module + ZeRO optimizer shard + lr scheduler state."""

import sys
import types
import uuid

import pytest
import torch


class FakeOptimizer:
    def __init__(self):
        self.state = {"m": torch.randn(4)}

    def state_dict(self):
        return {"state": {0: {"m": self.state["m"]}}}

    def load_state_dict(self, sd):
        self.state["m"] = sd["state"][0]["m"]


class FakeScheduler:
    def __init__(self):
        self.last_lr = 0.1

    def state_dict(self):
        return {"last_lr": self.last_lr}

    def load_state_dict(self, sd):
        self.last_lr = sd["last_lr"]


class FakeDeepSpeedEngine:
    """Shape of deepspeed.DeepSpeedEngine the integration touches."""

    def __init__(self):
        self.module = torch.nn.Linear(4, 2)
        self.optimizer = FakeOptimizer()
        self.lr_scheduler = FakeScheduler()
        self.config = {"zero_optimization": {"stage": 1}}


@pytest.mark.timeout(120)
def test_deepspeed_ckpt_roundtrip(tmp_path, monkeypatch):
    monkeypatch.setenv("ELASTIC_JOB_NAME", f"ds{uuid.uuid4().hex[:6]}")
    monkeypatch.setitem(sys.modules, "deepspeed", types.ModuleType("deepspeed"))
    from dlrover_amd.trainer.flash_checkpoint.deepspeed import (
        DeepSpeedCheckpointer,
    )

    eng = FakeDeepSpeedEngine()
    cp = DeepSpeedCheckpointer(eng, str(tmp_path / "ckpt"))
    before_w = eng.module.weight.detach().clone()
    before_m = eng.optimizer.state["m"].clone()
    cp.save_checkpoint(7)
    cp.wait_latest_checkpoint()

    with torch.no_grad():
        eng.module.weight.add_(1.0)
    eng.optimizer.state["m"] += 5.0
    eng.lr_scheduler.last_lr = 99.0

    sd = cp.load_checkpoint()
    assert sd is not None and sd["step"] == 7
    assert torch.equal(eng.module.weight.detach(), before_w)
    assert torch.equal(eng.optimizer.state["m"], before_m)
    assert eng.lr_scheduler.last_lr == 0.1
    assert sd["ds_config"]["zero_optimization"]["stage"] == 1
    cp.close()
    cp.engine.shm_handler.unlink()


def test_deepspeed_requires_library(tmp_path, monkeypatch):
    monkeypatch.delitem(sys.modules, "deepspeed", raising=False)
    from dlrover_amd.trainer.flash_checkpoint.deepspeed import (
        DeepSpeedCheckpointer,
    )

    with pytest.raises(ImportError):
        DeepSpeedCheckpointer(FakeDeepSpeedEngine(), str(tmp_path))
