#!/usr/bin/env python3
"""nanoGPT DDP training under dlrover-run — the CPU/gloo plumbing config
(BASELINE.json config #1) and the restart-on-kill demo.

Run:
    dlrover-run --standalone --nproc-per-node 2 examples/train_nanogpt.py \
        --steps 20 --ckpt-interval 5 --ckpt-dir /tmp/ng_ckpt

Fault injection (for tests): DLROVER_TEST_KILL_AT_STEP=<n> SIGKILLs rank 0 at
step n on the FIRST incarnation only; the elastic agent persists the shm
checkpoint, restarts workers, and training resumes from the last checkpoint.
"""

import argparse
import json
import os
import signal

import torch
import torch.distributed as dist
from torch.nn.parallel import DistributedDataParallel as DDP

from dlrover_amd.models import GPTConfig, NanoGPT
from dlrover_amd.trainer.elastic import ElasticTrainer
from dlrover_amd.trainer.flash_checkpoint import DdpCheckpointer, StorageType


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--batch", type=int, default=4)
    p.add_argument("--ckpt-interval", type=int, default=5)
    p.add_argument("--ckpt-dir", default="/tmp/dlrover_amd_ckpt/nanogpt")
    p.add_argument("--progress-file", default="")
    p.add_argument("--lr", type=float, default=3e-4)
    args = p.parse_args()

    use_gpu = torch.cuda.is_available()
    local_rank = int(os.getenv("LOCAL_RANK", "0"))
    if use_gpu:
        # modulo: N ranks can share one device (multi-rank RCCL validation
        # on a 1-GPU box; on the 8-GPU node the mapping is 1:1)
        local_rank = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(local_rank)
    # short collective timeout so a dead peer surfaces as a worker failure
    # quickly (elastic scale-down path); DLROVER_PG_TIMEOUT in seconds
    from datetime import timedelta

    pg_timeout = timedelta(seconds=int(os.getenv("DLROVER_PG_TIMEOUT", "1800")))
    dist.init_process_group(
        backend="nccl" if use_gpu else "gloo", timeout=pg_timeout
    )
    rank = dist.get_rank()
    device = torch.device(f"cuda:{local_rank}" if use_gpu else "cpu")

    torch.manual_seed(7)
    cfg = GPTConfig.tiny()
    model = NanoGPT(cfg).to(device)
    model = DDP(model)
    opt = torch.optim.AdamW(model.parameters(), lr=args.lr)

    trainer = ElasticTrainer(model)  # reports global step for the monitor
    cp = DdpCheckpointer(args.ckpt_dir, model, opt)
    start_step = 0
    restored = cp.load_checkpoint()
    if restored is not None:
        start_step = int(restored.get("step", 0))
        if rank == 0:
            print(f"[train] resumed from checkpoint step {start_step}", flush=True)

    kill_at = int(os.getenv("DLROVER_TEST_KILL_AT_STEP", "0"))
    hang_at = int(os.getenv("DLROVER_TEST_HANG_AT_STEP", "0"))
    incarnation = int(os.getenv("TORCHELASTIC_RESTART_COUNT", "0"))

    torch.manual_seed(1 + rank)
    ids = torch.randint(0, cfg.vocab_size, (args.batch, cfg.block_size), device=device)

    for step in range(start_step + 1, args.steps + 1):
        if hang_at and step == hang_at and incarnation == 0 and rank == 0:
            # simulate a wedged collective: this rank stops participating
            # (config #5 — the master's hang diagnostician must recover us)
            print(f"[train] injecting hang at step {step}", flush=True)
            import time as _t

            _t.sleep(3600)
        if (os.getenv("DLROVER_TEST_TIMELINE_FLAG") == "1"
                and step == max(2, args.steps * 3 // 5) and rank == 0):
            # ask the preloaded hiptimer to dump its kernel-trace ring
            mdir = os.getenv("HIPTIMER_METRICS_DIR", "/tmp/hiptimer")
            open(os.path.join(mdir, f"dump_timeline_{rank}"), "w").close()
        loss = model(ids, ids.clone())
        opt.zero_grad()
        loss.backward()
        opt.step()
        trainer.global_step = step
        trainer._report_step()
        if step % args.ckpt_interval == 0:
            sd = {
                "step": step,
                "model": model.module.state_dict(),
                "optimizer": opt.state_dict(),
            }
            cp.save_checkpoint(step, state_dict=sd, storage_type=StorageType.DISK)
        if args.progress_file and rank == 0:
            with open(args.progress_file, "a") as f:
                f.write(
                    json.dumps(
                        {"step": step, "loss": round(loss.item(), 4),
                         "incarnation": incarnation, "resumed_from": start_step,
                         "world": dist.get_world_size(),
                         "device": str(device)}
                    )
                    + "\n"
                )
        if kill_at and step == kill_at and incarnation == 0 and rank == 0:
            print(f"[train] injecting SIGKILL at step {step}", flush=True)
            os.kill(os.getpid(), signal.SIGKILL)

    cp.wait_latest_checkpoint()
    if rank == 0:
        print(f"[train] done at step {args.steps}, loss={loss.item():.4f}", flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
