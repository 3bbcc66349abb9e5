#!/usr/bin/env python3
"""Llama-3 FSDP bf16 training under dlrover-run with flash checkpoint —
BASELINE config #2 ("Llama-3 8B FSDP bf16 on 8xMI355X, flash-checkpoint every
50 steps + 1 injected SIGKILL").

Run on one 8-GPU MI355X node:
    dlrover-run --standalone --nproc-per-node 8 examples/train_llama_fsdp.py \
        --model llama3_8b --steps 200 --ckpt-interval 50

Fault injection: DLROVER_TEST_KILL_AT_STEP=<n> SIGKILLs rank 0 at step n on
the first incarnation; the agent persists the last shm snapshot, restarts
workers, and training resumes from the latest committed checkpoint.
"""

import argparse
import json
import os
import signal
import time

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3_8b",
                   choices=["llama3_8b", "small_1b", "tiny"])
    p.add_argument("--steps", type=int, default=100)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--seq", type=int, default=4096)
    p.add_argument("--ckpt-interval", type=int, default=50)
    p.add_argument("--ckpt-dir", default="/tmp/dlrover_amd_ckpt/llama")
    p.add_argument("--progress-file", default="")
    p.add_argument("--lr", type=float, default=1e-4)
    p.add_argument("--act-ckpt", action="store_true",
                   help="recompute transformer blocks in backward")
    args = p.parse_args()

    from dlrover_amd.models import LlamaConfig, LlamaForCausalLM
    from dlrover_amd.ops import FusedAdamW
    from dlrover_amd.trainer.elastic import ElasticTrainer
    from dlrover_amd.trainer.flash_checkpoint import (
        FsdpShardCheckpointer,
        StorageType,
    )

    use_gpu = torch.cuda.is_available()
    local_rank = int(os.getenv("LOCAL_RANK", "0"))
    if use_gpu:
        torch.cuda.set_device(local_rank)
        from dlrover_amd.ops.api import hip_ops

        hip_ops()  # fail loudly without the native extension
    dist.init_process_group("nccl" if use_gpu else "gloo")
    rank = dist.get_rank()
    device = torch.device(f"cuda:{local_rank}" if use_gpu else "cpu")

    torch.manual_seed(0)
    cfg = (
        LlamaConfig.llama3_8b(max_seq_len=max(args.seq, 4096))
        if args.model == "llama3_8b"
        else getattr(LlamaConfig, args.model)()
    )
    if args.model == "tiny":
        args.seq = min(args.seq, cfg.max_seq_len)
    cfg.activation_checkpointing = args.act_ckpt
    with device:
        model = LlamaForCausalLM(cfg)
    model = model.to(device)
    use_fsdp = use_gpu or os.getenv("DLROVER_FSDP_CPU", "") == "1"
    if use_gpu:
        model = model.bfloat16()
    if use_fsdp:
        # fully_shard even at world 1: DTensor shard0 tags keep checkpoints
        # UCP-reshardable in BOTH scale directions (grow needs the ws-1 ckpt
        # tagged shard0, not replicated)
        from torch.distributed.fsdp import fully_shard

        for blk in model.blocks:
            fully_shard(blk)
        fully_shard(model)
    opt = FusedAdamW(model.parameters(), lr=args.lr, weight_decay=0.1)
    trainer = ElasticTrainer(model)

    cp = FsdpShardCheckpointer(args.ckpt_dir, model, opt)
    start_step = 0
    restored = cp.load_checkpoint()
    if restored is not None:
        start_step = int(restored.get("step", 0))
        if rank == 0:
            print(f"[train] resumed from step {start_step}", flush=True)

    kill_at = int(os.getenv("DLROVER_TEST_KILL_AT_STEP", "0"))
    incarnation = int(os.getenv("TORCHELASTIC_RESTART_COUNT", "0"))

    torch.manual_seed(1 + rank)
    ids = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), device=device)
    labels = torch.randint(0, cfg.vocab_size, (args.batch, args.seq), device=device)

    for step in range(start_step + 1, args.steps + 1):
        t0 = time.perf_counter()
        with trainer.step():
            loss = model(ids, labels)
            loss.backward()
        opt.step()
        opt.zero_grad()
        if step % args.ckpt_interval == 0:
            blocking = cp.save_checkpoint(step, storage_type=StorageType.DISK)
            if rank == 0:
                print(f"[train] ckpt@{step} blocked {blocking * 1e3:.1f} ms",
                      flush=True)
        if args.progress_file and rank == 0:
            with open(args.progress_file, "a") as f:
                f.write(json.dumps({
                    "step": step, "loss": round(float(loss.item()), 4),
                    "step_s": round(time.perf_counter() - t0, 3),
                    "ts": round(time.time(), 3),
                    "incarnation": incarnation, "resumed_from": start_step,
                    "world": dist.get_world_size(),
                    "device": str(device),
                }) + "\n")
        if kill_at and step == kill_at and incarnation == 0 and rank == 0:
            print(f"[train] injecting SIGKILL at step {step}", flush=True)
            os.kill(os.getpid(), signal.SIGKILL)

    cp.wait_latest_checkpoint()
    if rank == 0:
        print(f"[train] done at step {args.steps} loss={loss.item():.4f}",
              flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
