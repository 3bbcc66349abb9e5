"""UCP: resume an FSDP local-shard checkpoint at a DIFFERENT world size
(elastic 2 -> 1 here; the mechanism is world-size agnostic)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from dlrover_amd.common.global_context import find_free_port


def _phase_a_worker(rank, port, tmpdir, results):
    os.environ.update(
        {
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": "2",
            "ELASTIC_JOB_NAME": f"ucpa{port}",
            "DLROVER_IPC_SOCKET_DIR": os.path.join(tmpdir, "ipc"),
        }
    )
    dist.init_process_group("gloo", rank=rank, world_size=2)
    try:
        from torch.distributed.fsdp import fully_shard

        from dlrover_amd.models import LlamaConfig, LlamaForCausalLM
        from dlrover_amd.ops import FusedAdamW
        from dlrover_amd.trainer.flash_checkpoint import (
            FsdpShardCheckpointer,
            StorageType,
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
        cp = FsdpShardCheckpointer(os.path.join(tmpdir, "ckpt"), model, opt)
        cp.save_checkpoint(9, storage_type=StorageType.DISK)
        cp.wait_latest_checkpoint()
        dist.barrier()
        cp.close()
        cp.engine.shm_handler.unlink()
        results[f"a{rank}"] = "ok"
    except Exception as e:  # noqa: BLE001
        import traceback

        results[f"a{rank}"] = f"FAIL {e}\n{traceback.format_exc()}"
        raise
    finally:
        dist.destroy_process_group()


def _phase_b_worker(rank, port, tmpdir, results):
    os.environ.update(
        {
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": "0",
            "LOCAL_RANK": "0",
            "WORLD_SIZE": "1",
            "ELASTIC_JOB_NAME": f"ucpb{port}",
            "DLROVER_IPC_SOCKET_DIR": os.path.join(tmpdir, "ipcb"),
        }
    )
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        from torch.distributed.fsdp import fully_shard

        from dlrover_amd.models import LlamaConfig, LlamaForCausalLM
        from dlrover_amd.ops import FusedAdamW
        from dlrover_amd.trainer.flash_checkpoint import FsdpShardCheckpointer

        torch.manual_seed(123)  # different init: restore must overwrite it
        cfg = LlamaConfig.tiny()
        model = LlamaForCausalLM(cfg)
        for blk in model.blocks:
            fully_shard(blk)
        fully_shard(model)
        opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.0)
        # optimizer state must exist before UCP can fill it
        ids = torch.randint(0, cfg.vocab_size, (1, 8))
        model(ids, ids.clone()).backward()
        opt.step()
        opt.zero_grad()

        cp = FsdpShardCheckpointer(os.path.join(tmpdir, "ckpt"), model, opt)
        out = cp.load_checkpoint()
        assert out is not None and out.get("step") == 9, out

        # at ws=1 each param's local shard IS the full tensor: compare with
        # the concatenation of the phase-A shard files
        shard0 = torch.load(
            os.path.join(tmpdir, "ckpt", "9", "rank_00000.pt"), weights_only=False
        )
        shard1 = torch.load(
            os.path.join(tmpdir, "ckpt", "9", "rank_00001.pt"), weights_only=False
        )
        name, p = next(iter(model.named_parameters()))
        expected = torch.cat([shard0["model"][name], shard1["model"][name]], dim=0)
        torch.testing.assert_close(p.to_local(), expected)

        # optimizer master weights resharded too
        st = opt.state[next(iter(opt.param_groups[0]["params"]))]
        exp_m = torch.cat(
            [shard0["optimizer"]["state"][0]["master_param"],
             shard1["optimizer"]["state"][0]["master_param"]], dim=0
        )
        torch.testing.assert_close(st["master_param"], exp_m)

        # training continues
        loss = model(ids, ids.clone())
        loss.backward()
        opt.step()
        cp.close()
        cp.engine.shm_handler.unlink()
        results["b"] = "ok"
    except Exception as e:  # noqa: BLE001
        import traceback

        results["b"] = f"FAIL {e}\n{traceback.format_exc()}"
        raise
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(420)
def test_ucp_reshard_2_to_1(tmp_path):
    ctx = mp.get_context("spawn")
    with mp.Manager() as mgr:
        results = mgr.dict()
        port = find_free_port()
        procs = [
            ctx.Process(target=_phase_a_worker, args=(r, port, str(tmp_path), results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=180)
        assert results.get("a0") == "ok" and results.get("a1") == "ok", dict(results)

        port2 = find_free_port()
        pb = ctx.Process(target=_phase_b_worker, args=(0, port2, str(tmp_path), results))
        pb.start()
        pb.join(timeout=180)
        assert results.get("b") == "ok", dict(results)


def _phase_c_save_ws1(rank, port, tmpdir, results):
    os.environ.update(
        {
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": "0",
            "LOCAL_RANK": "0",
            "WORLD_SIZE": "1",
            "ELASTIC_JOB_NAME": f"ucpc{port}",
            "DLROVER_IPC_SOCKET_DIR": os.path.join(tmpdir, "ipcc"),
        }
    )
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        from torch.distributed.fsdp import fully_shard

        from dlrover_amd.models import LlamaConfig, LlamaForCausalLM
        from dlrover_amd.ops import FusedAdamW
        from dlrover_amd.trainer.flash_checkpoint import (
            FsdpShardCheckpointer,
            StorageType,
        )

        torch.manual_seed(0)
        cfg = LlamaConfig.tiny()
        model = LlamaForCausalLM(cfg)
        for blk in model.blocks:
            fully_shard(blk)
        fully_shard(model)
        opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.0)
        ids = torch.randint(0, cfg.vocab_size, (2, 16))
        model(ids, ids.clone()).backward()
        opt.step()
        opt.zero_grad()
        cp = FsdpShardCheckpointer(os.path.join(tmpdir, "ckpt_g"), model, opt)
        cp.save_checkpoint(11, storage_type=StorageType.DISK)
        cp.wait_latest_checkpoint()
        cp.close()
        cp.engine.shm_handler.unlink()
        results["c"] = "ok"
    except Exception as e:  # noqa: BLE001
        import traceback

        results["c"] = f"FAIL {e}\n{traceback.format_exc()}"
        raise
    finally:
        dist.destroy_process_group()


def _phase_d_load_ws2(rank, port, tmpdir, results):
    os.environ.update(
        {
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": "2",
            "ELASTIC_JOB_NAME": f"ucpd{port}",
            "DLROVER_IPC_SOCKET_DIR": os.path.join(tmpdir, "ipcd"),
        }
    )
    dist.init_process_group("gloo", rank=rank, world_size=2)
    try:
        from torch.distributed.fsdp import fully_shard

        from dlrover_amd.models import LlamaConfig, LlamaForCausalLM
        from dlrover_amd.ops import FusedAdamW
        from dlrover_amd.trainer.flash_checkpoint import FsdpShardCheckpointer

        torch.manual_seed(77 + rank)
        cfg = LlamaConfig.tiny()
        model = LlamaForCausalLM(cfg)
        for blk in model.blocks:
            fully_shard(blk)
        fully_shard(model)
        opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.0)
        ids = torch.randint(0, cfg.vocab_size, (1, 8))
        model(ids, ids.clone()).backward()
        opt.step()
        opt.zero_grad()

        cp = FsdpShardCheckpointer(os.path.join(tmpdir, "ckpt_g"), model, opt)
        out = cp.load_checkpoint()
        assert out is not None and out.get("step") == 11, out

        # this rank's resharded slice must equal the matching rows of the
        # ws=1 full tensor (offsets = gathered local row counts)
        full = torch.load(
            os.path.join(tmpdir, "ckpt_g", "11", "rank_00000.pt"),
            weights_only=False,
        )
        name, p = next(iter(model.named_parameters()))
        rows = torch.tensor([p.to_local().shape[0]])
        gathered = [torch.zeros_like(rows) for _ in range(2)]
        dist.all_gather(gathered, rows)
        off = int(gathered[0].item()) * rank
        want = full["model"][name][off : off + int(rows.item())]
        torch.testing.assert_close(p.to_local(), want)

        # training continues at the new world size
        model(ids, ids.clone()).backward()
        opt.step()
        dist.barrier()
        cp.close()
        cp.engine.shm_handler.unlink()
        results[f"d{rank}"] = "ok"
    except Exception as e:  # noqa: BLE001
        import traceback

        results[f"d{rank}"] = f"FAIL {e}\n{traceback.format_exc()}"
        raise
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(420)
def test_ucp_reshard_1_to_2(tmp_path):
    ctx = mp.get_context("spawn")
    with mp.Manager() as mgr:
        results = mgr.dict()
        port = find_free_port()
        pc = ctx.Process(target=_phase_c_save_ws1, args=(0, port, str(tmp_path), results))
        pc.start()
        pc.join(timeout=180)
        assert results.get("c") == "ok", dict(results)

        port2 = find_free_port()
        procs = [
            ctx.Process(target=_phase_d_load_ws2, args=(r, port2, str(tmp_path), results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=180)
        assert results.get("d0") == "ok" and results.get("d1") == "ok", dict(results)
