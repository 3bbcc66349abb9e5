"""FSDP2 sharded flash checkpoint over a real 2-process gloo group (CPU).

Covers the sharded engine save/restore path the 8-GPU bench uses — on CPU so
the driver's no-GPU lane exercises it (BASELINE config #2's plumbing).
"""

import os
import time

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from dlrover_amd.common.global_context import find_free_port

WS = 2


def _worker(rank, port, tmpdir, results):
    os.environ.update(
        {
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": str(WS),
            "ELASTIC_JOB_NAME": f"fsdp{port}",
            "DLROVER_IPC_SOCKET_DIR": os.path.join(tmpdir, "ipc"),
        }
    )
    dist.init_process_group("gloo", rank=rank, world_size=WS)
    try:
        from torch.distributed.fsdp import fully_shard

        from dlrover_amd.models import LlamaConfig, LlamaForCausalLM
        from dlrover_amd.ops import FusedAdamW
        from dlrover_amd.trainer.flash_checkpoint.engine import (
            ShardedCheckpointEngine,
        )

        torch.manual_seed(0)
        cfg = LlamaConfig.tiny()
        model = LlamaForCausalLM(cfg)
        for blk in model.blocks:
            fully_shard(blk)
        fully_shard(model)
        opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.0)

        ids = torch.randint(0, cfg.vocab_size, (2, 16))
        loss = model(ids, ids.clone())
        loss.backward()
        opt.step()
        opt.zero_grad()

        engine = ShardedCheckpointEngine(os.path.join(tmpdir, "ckpt"))
        sd = engine.gather_state_dict(model, opt)
        sd["step"] = 3
        engine.save_to_memory(3, sd)

        # snapshot a reference param shard, then perturb the live model
        name0, p0 = next(iter(model.named_parameters()))
        ref = p0.to_local().clone()
        with torch.no_grad():
            for p in model.parameters():
                p.to_local().add_(1.0)
        assert not torch.allclose(p0.to_local(), ref)

        out = engine.restore_into(model, opt)
        assert out is not None and out["step"] == 3
        assert torch.allclose(p0.to_local(), ref)

        # training still works after restore
        loss2 = model(ids, ids.clone())
        loss2.backward()
        opt.step()

        engine.close()
        engine.shm_handler.unlink()
        results[rank] = "ok"
    except Exception as e:  # noqa: BLE001
        import traceback

        results[rank] = f"FAIL rank{rank}: {e}\n{traceback.format_exc()}"
        raise
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_fsdp2_sharded_ckpt_roundtrip(tmp_path):
    port = find_free_port()
    ctx = mp.get_context("spawn")
    with mp.Manager() as mgr:
        results = mgr.dict()
        procs = [
            ctx.Process(target=_worker, args=(r, port, str(tmp_path), results))
            for r in range(WS)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=240)
        outcomes = dict(results)
    assert all(outcomes.get(r) == "ok" for r in range(WS)), outcomes
