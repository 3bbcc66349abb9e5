#!/usr/bin/env python3
"""Llama TP x PP training under dlrover-run with the Megatron-layout flash
checkpoint — BASELINE config #3 ("Llama-3 8B Megatron TP=2 PP=2 checkpoint
engine on 8xMI355X").

8 GPUs = TP2 x PP2 x DP2:
    dlrover-run --standalone --nproc-per-node 8 examples/train_llama_tp_pp.py \
        --model llama3_8b --tp 2 --pp 2 --steps 200 --ckpt-interval 50

CPU plumbing (4 gloo ranks, tiny model):
    torchrun --standalone --nproc-per-node 4 examples/train_llama_tp_pp.py \
        --model tiny --tp 2 --pp 2 --steps 4 --seq 16
"""

import argparse
import json
import os
import time

import torch
import torch.distributed as dist

# runnable under bare torchrun from anywhere: the package lives in-tree
sys_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
import sys  # noqa: E402

if sys_root not in sys.path:
    sys.path.insert(0, sys_root)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3_8b",
                   choices=["llama3_8b", "small_1b", "tiny"])
    p.add_argument("--tp", type=int, default=2)
    p.add_argument("--pp", type=int, default=2)
    p.add_argument("--steps", type=int, default=100)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--seq", type=int, default=4096)
    p.add_argument("--micro-batches", type=int, default=4)
    p.add_argument("--ckpt-interval", type=int, default=50)
    p.add_argument("--ckpt-dir", default="/tmp/dlrover_amd_ckpt/llama_tp_pp")
    p.add_argument("--progress-file", default="")
    p.add_argument("--lr", type=float, default=1e-4)
    args = p.parse_args()

    use_gpu = torch.cuda.is_available()
    local_rank = int(os.getenv("LOCAL_RANK", "0"))
    if use_gpu:
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        from dlrover_amd.ops.api import hip_ops

        hip_ops()
    dist.init_process_group("nccl" if use_gpu else "gloo")
    world = dist.get_world_size()
    dp = world // (args.tp * args.pp)
    if dp * args.tp * args.pp != world:
        raise SystemExit(f"world {world} != tp{args.tp} x pp{args.pp} x dp{dp}")

    from dlrover_amd.models import LlamaConfig
    from dlrover_amd.models.llama_parallel import LlamaStage
    from dlrover_amd.ops import FusedAdamW
    from dlrover_amd.parallel.pgroups import ParallelDims, ParallelGroups
    from dlrover_amd.parallel.pp import PipelineRunner
    from dlrover_amd.trainer.flash_checkpoint.checkpointer import StorageType
    from dlrover_amd.trainer.flash_checkpoint.megatron import (
        MegatronCheckpointer,
    )

    os.environ.setdefault("ELASTIC_JOB_NAME",
                          f"tp{args.tp}pp{args.pp}dp{dp}_{args.model}")
    groups = ParallelGroups(ParallelDims(tp=args.tp, pp=args.pp, dp=dp))
    torch.manual_seed(0)
    cfg = (
        LlamaConfig.llama3_8b(max_seq_len=max(args.seq, 4096))
        if args.model == "llama3_8b"
        else getattr(LlamaConfig, args.model)()
    )
    if args.model == "tiny":
        args.seq = min(args.seq, cfg.max_seq_len)
    device = torch.device(
        f"cuda:{local_rank % torch.cuda.device_count()}" if use_gpu else "cpu"
    )
    with device:
        stage = LlamaStage(cfg, groups)
    stage = stage.to(device)
    if use_gpu:
        stage = stage.bfloat16()
    opt = FusedAdamW(stage.parameters(), lr=args.lr, weight_decay=0.1)
    runner = PipelineRunner(stage, groups, cfg.hidden_size)
    cp = MegatronCheckpointer(args.ckpt_dir, groups, stage, opt)

    start_step = 0
    try:
        restored = cp.load_checkpoint()
        if restored is not None:
            start_step = int(restored.get("step", 0))
            if rank_zero := (dist.get_rank() == 0):
                print(f"[tp_pp] resumed from step {start_step}", flush=True)
            del rank_zero
    except (RuntimeError, KeyError) as e:
        # foreign/incompatible snapshot (different geometry or model):
        # start fresh rather than bricking the job
        print(f"[tp_pp] ignoring incompatible checkpoint: {e}", flush=True)

    torch.manual_seed(1 + groups.dp_rank)
    rank = dist.get_rank()
    for step in range(start_step + 1, args.steps + 1):
        t0 = time.perf_counter()
        micros = [
            torch.randint(0, cfg.vocab_size, (args.batch, args.seq),
                          device=device)
            for _ in range(args.micro_batches)
        ]
        loss = runner.train_step(micros, [m.clone() for m in micros],
                                 schedule="1f1b")
        opt.step()
        opt.zero_grad()
        if step % args.ckpt_interval == 0:
            cp.save_checkpoint(step, storage_type=StorageType.DISK)
        writes_progress = (groups.is_last_stage and groups.tp_rank == 0
                           and groups.dp_rank == 0)  # the rank holding loss
        if args.progress_file and writes_progress:
            with open(args.progress_file, "a") as f:
                f.write(json.dumps({
                    "step": step,
                    "loss": float(loss.item()) if loss is not None else None,
                    "step_s": round(time.perf_counter() - t0, 3),
                    "tp": args.tp, "pp": args.pp, "dp": dp,
                }) + "\n")
    cp.wait_latest_checkpoint()
    if rank == 0:
        print(f"[tp_pp] done at step {args.steps}", flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
