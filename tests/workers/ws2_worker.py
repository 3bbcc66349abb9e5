"""Worker for multi-rank-RCCL-on-one-GPU tests (spawned by
tests/test_multirank_gpu.py with RANK/WORLD_SIZE env set, all ranks on
cuda:0). Modes:
  allreduce  — RCCL all_reduce sanity
  fsdp       — FSDP2 fully_shard tiny Llama + 2 train steps + flash ckpt
               save-to-shm / perturb / restore_into roundtrip
Exits 0 on success; prints MARK lines the parent asserts on.
"""

import os
import sys

import torch
import torch.distributed as dist


def main():
    mode = sys.argv[1]
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    # distinct devices when available (e.g. CPX partition: 1 MI355X -> 8
    # logical XCD devices); RCCL refuses two ranks on ONE device
    dev = torch.device(f"cuda:{rank % torch.cuda.device_count()}")
    torch.cuda.set_device(dev)
    dist.init_process_group("nccl", rank=rank, world_size=world,
                            device_id=dev)
    if mode == "allreduce":
        t = torch.full((1 << 20,), float(rank + 1), device=dev)
        dist.all_reduce(t)
        expect = world * (world + 1) / 2
        assert torch.all(t == expect), t[:4]
        # reduce_scatter + all_gather (the FSDP2 collective pair)
        outs = [torch.empty(1 << 18, device=dev) for _ in range(world)]
        dist.all_gather(outs, t[: 1 << 18])
        print(f"MARK allreduce ok rank={rank}", flush=True)
    elif mode == "fsdp":
        from torch.distributed.fsdp import fully_shard

        from dlrover_amd.models import LlamaConfig, LlamaForCausalLM
        from dlrover_amd.ops import FusedAdamW
        from dlrover_amd.trainer.flash_checkpoint import FsdpShardCheckpointer

        torch.manual_seed(0)
        cfg = LlamaConfig.tiny()
        with dev:
            model = LlamaForCausalLM(cfg)
        model = model.bfloat16()
        for blk in model.blocks:
            fully_shard(blk)
        fully_shard(model)
        opt = FusedAdamW(model.parameters(), lr=1e-3)
        ids = torch.randint(0, cfg.vocab_size, (2, 64), device=dev)
        labels = torch.randint(0, cfg.vocab_size, (2, 64), device=dev)
        for _ in range(2):
            loss = model(ids, labels)
            loss.backward()
            opt.step()
            opt.zero_grad()
        cp = FsdpShardCheckpointer(os.environ["CKPT_DIR"], model, opt)
        sd = cp.engine.gather_state_dict(model, opt)
        sd["step"] = 2
        cp.engine.save_to_memory(2, sd)
        cp.engine.wait_saving()
        # perturb, then restore must bring the exact values back
        before = {
            n: t.to_local().clone() if hasattr(t, "to_local") else t.clone()
            for n, t in model.named_parameters()
        }
        with torch.no_grad():
            for p in model.parameters():
                local = p.to_local() if hasattr(p, "to_local") else p
                local.add_(1.0)
        restored = cp.engine.restore_into(model, opt)
        assert restored is not None
        for n, t in model.named_parameters():
            local = t.to_local() if hasattr(t, "to_local") else t
            assert torch.equal(local, before[n]), n
        loss2 = model(ids, labels)  # still trains after restore
        loss2.backward()
        opt.step()
        cp.close()
        print(f"MARK fsdp ok rank={rank} loss={loss2.item():.3f}", flush=True)
    else:
        raise SystemExit(f"unknown mode {mode}")
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
