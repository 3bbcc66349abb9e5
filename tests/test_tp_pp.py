"""TP=2 x PP=2 Llama over 4 gloo processes: numerics vs the single-process
model, GPipe training step, and the Megatron-style checkpoint engine
(BASELINE config #3 plumbing on CPU)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from dlrover_amd.common.global_context import find_free_port

WS = 4


def _seed_stage_from_full(stage, full_model, groups, cfg):
    """Copy the single-process model's weights into this rank's TP/PP shards."""
    from dlrover_amd.parallel.tp import shard_full_weight

    tp, tr = groups.dims.tp, groups.tp_rank
    lo, hi = stage.layer_range
    with torch.no_grad():
        if stage.embed is not None:
            stage.embed.weight.copy_(full_model.embed.weight)
        if stage.final_norm is not None:
            stage.final_norm.weight.copy_(full_model.final_norm.weight)
            stage.lm_head.weight.copy_(full_model.lm_head.weight)
        hd = cfg.head_dim
        nh, nkv = cfg.n_heads, cfg.n_kv_heads
        for li, blk in enumerate(stage.blocks):
            src = full_model.blocks[lo + li]
            blk.attn_norm.weight.copy_(src.attn_norm.weight)
            blk.mlp_norm.weight.copy_(src.mlp_norm.weight)
            # qkv: rows [q | k | v]; shard each section's heads
            w = src.attn.qkv_proj.weight
            qw, kw, vw = w.split([nh * hd, nkv * hd, nkv * hd], dim=0)
            blk.attn.qkv_proj.weight.copy_(
                torch.cat(
                    [
                        shard_full_weight(qw, tr, tp, 0),
                        shard_full_weight(kw, tr, tp, 0),
                        shard_full_weight(vw, tr, tp, 0),
                    ],
                    dim=0,
                )
            )
            blk.attn.o_proj.weight.copy_(
                shard_full_weight(src.attn.o_proj.weight, tr, tp, 1)
            )
            gw, uw = src.mlp.gate_up_proj.weight.chunk(2, dim=0)
            blk.mlp.gate_up_proj.weight.copy_(
                torch.cat(
                    [shard_full_weight(gw, tr, tp, 0), shard_full_weight(uw, tr, tp, 0)],
                    dim=0,
                )
            )
            blk.mlp.down_proj.weight.copy_(
                shard_full_weight(src.mlp.down_proj.weight, tr, tp, 1)
            )


def _worker(rank, port, tmpdir, results):
    os.environ.update(
        {
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": str(WS),
            "ELASTIC_JOB_NAME": f"tppp{port}",
            "DLROVER_IPC_SOCKET_DIR": os.path.join(tmpdir, "ipc"),
        }
    )
    dist.init_process_group("gloo", rank=rank, world_size=WS)
    try:
        from dlrover_amd.models.llama import LlamaConfig, LlamaForCausalLM
        from dlrover_amd.models.llama_parallel import LlamaStage
        from dlrover_amd.parallel.pgroups import ParallelDims, ParallelGroups
        from dlrover_amd.parallel.pp import PipelineRunner
        from dlrover_amd.trainer.flash_checkpoint.megatron import (
            MegatronCheckpointer,
        )
        from dlrover_amd.trainer.flash_checkpoint.checkpointer import StorageType

        groups = ParallelGroups(ParallelDims(tp=2, pp=2, dp=1))
        torch.manual_seed(0)
        cfg = LlamaConfig.tiny()
        full = LlamaForCausalLM(cfg)  # identical on all ranks (seed 0)
        stage = LlamaStage(cfg, groups)
        _seed_stage_from_full(stage, full, groups, cfg)

        torch.manual_seed(42)
        B, S = 2, 16
        ids = torch.randint(0, cfg.vocab_size, (B, S))
        labels = ids.clone()

        # ---- forward equivalence: pipeline loss == single-process loss ----
        runner = PipelineRunner(stage, groups, cfg.hidden_size)
        loss = runner.train_step([ids], [labels])
        if groups.is_last_stage:
            ref_loss = full(ids, labels)
            assert torch.allclose(loss, ref_loss.detach(), rtol=1e-3, atol=1e-4), (
                loss,
                ref_loss,
            )

        # grads must exist on every stage after the pipeline backward
        grads = [p.grad for p in stage.parameters()]
        assert all(g is not None for g in grads), "missing grads on a stage"

        # ---- 1F1B and GPipe must produce the same gradients (4 microbatches;
        # the schedules only reorder the backward accumulation) ----
        micros = [torch.randint(0, cfg.vocab_size, (B, S)) for _ in range(4)]
        mlabels = [m.clone() for m in micros]
        for p in stage.parameters():
            p.grad = None
        loss_g = runner.train_step(micros, mlabels, schedule="gpipe")
        grads_g = [p.grad.detach().clone() for p in stage.parameters()]
        for p in stage.parameters():
            p.grad = None
        loss_1 = runner.train_step(micros, mlabels, schedule="1f1b")
        if groups.is_last_stage:
            assert torch.allclose(loss_1, loss_g, rtol=1e-5, atol=1e-6)
        for g1, gg in zip([p.grad for p in stage.parameters()], grads_g):
            assert torch.allclose(g1, gg, rtol=1e-4, atol=1e-6)

        # ---- Megatron-style checkpoint roundtrip ----
        opt = torch.optim.AdamW(stage.parameters(), lr=1e-4)
        opt.step()
        cp = MegatronCheckpointer(os.path.join(tmpdir, "ckpt"), groups, stage, opt)
        cp.save_checkpoint(7, storage_type=StorageType.DISK)
        cp.wait_latest_checkpoint()
        dist.barrier()
        if rank == 0:
            base = os.path.join(tmpdir, "ckpt", "7")
            names = sorted(os.listdir(base))
            shards = [n for n in names if n.startswith("mp_rank_")]
            assert shards == [
                "mp_rank_00_000.pt",
                "mp_rank_00_001.pt",
                "mp_rank_01_000.pt",
                "mp_rank_01_001.pt",
            ], shards

        ref_param = next(iter(stage.parameters())).detach().clone()
        with torch.no_grad():
            for p in stage.parameters():
                p.add_(1.0)
        out = cp.load_checkpoint()
        assert out is not None and out["step"] == 7
        assert torch.allclose(next(iter(stage.parameters())), ref_param)

        cp.close()
        cp.engine.shm_handler.unlink()
        results[rank] = "ok"
    except Exception as e:  # noqa: BLE001
        import traceback

        results[rank] = f"FAIL rank{rank}: {e}\n{traceback.format_exc()}"
        raise
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(420)
def test_tp2_pp2_llama_and_megatron_ckpt(tmp_path):
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
            p.join(timeout=360)
        outcomes = dict(results)
    assert all(outcomes.get(r) == "ok" for r in range(WS)), outcomes


@pytest.mark.timeout(420)
def test_tp_pp_example_e2e(tmp_path):
    """The user-facing TP x PP example trains, checkpoints and resumes on
    4 gloo ranks (config #3 entry point)."""
    import json
    import subprocess
    import sys
    import uuid

    progress = tmp_path / "prog.jsonl"
    env = dict(os.environ)
    env.update({
        "ELASTIC_JOB_NAME": f"tpex{uuid.uuid4().hex[:6]}",
        "DLROVER_IPC_SOCKET_DIR": str(tmp_path / "ipc"),
        "MASTER_ADDR": "127.0.0.1",
    })
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--standalone",
        "--master-addr", "127.0.0.1", "--nproc-per-node", "4",
        os.path.join(root, "examples", "train_llama_tp_pp.py"),
        "--model", "tiny", "--tp", "2", "--pp", "2", "--steps", "4",
        "--seq", "16", "--ckpt-interval", "2",
        "--ckpt-dir", str(tmp_path / "ckpt"),
        "--progress-file", str(progress),
    ]
    out = subprocess.run(cmd, cwd=root, env=env, capture_output=True,
                         text=True, timeout=360)
    assert out.returncode == 0, out.stderr[-4000:]
    rows = [json.loads(l) for l in progress.read_text().splitlines()]
    assert rows[-1]["step"] == 4 and rows[-1]["loss"] is not None
    # resume: a second run continues from the committed step
    cmd[cmd.index("--steps") + 1] = "6"
    out2 = subprocess.run(cmd, cwd=root, env=env, capture_output=True,
                          text=True, timeout=360)
    assert out2.returncode == 0, out2.stderr[-4000:]
    rows = [json.loads(l) for l in progress.read_text().splitlines()]
    steps = [r["step"] for r in rows]
    assert steps[-1] == 6 and 5 in steps and steps.count(4) == 1, steps


@pytest.mark.timeout(420)
def test_tp1_pp2_example_does_not_world_allreduce(tmp_path):
    """tp=1 with pp=2 must not fall through to whole-world collectives
    (group None == degenerate): the example trains cleanly at tp1 pp2."""
    import subprocess
    import sys
    import uuid

    env = dict(os.environ)
    env.update({
        "ELASTIC_JOB_NAME": f"tp1{uuid.uuid4().hex[:6]}",
        "DLROVER_IPC_SOCKET_DIR": str(tmp_path / "ipc"),
        "MASTER_ADDR": "127.0.0.1",
    })
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--standalone",
         "--master-addr", "127.0.0.1", "--nproc-per-node", "2",
         os.path.join(root, "examples", "train_llama_tp_pp.py"),
         "--model", "tiny", "--tp", "1", "--pp", "2", "--steps", "3",
         "--seq", "16", "--ckpt-interval", "2",
         "--ckpt-dir", str(tmp_path / "ckpt")],
        cwd=root, env=env, capture_output=True, text=True, timeout=360,
    )
    assert out.returncode == 0, out.stderr[-4000:]
