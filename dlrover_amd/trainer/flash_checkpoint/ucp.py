"""Universal checkpoint (UCP): reshard an FSDP local-shard checkpoint across
a world-size change (ref: training.py:1548-1651 UCP hook + UcpRdzvManager —
the reference converts Megatron/DeepSpeed ckpts; ours reshapes OUR sharded
layout).

FSDP2 (fully_shard) places every parameter as DTensor Shard(0): rank r holds
a contiguous dim-0 slice, torch.chunk boundaries. The sharded engine stores
each rank's local tensors plus a per-key sharding tag. To resume at a NEW
world size: concatenate every old rank's slice back to the full tensor, then
cut this rank's new slice — done streaming per old shard file so peak memory
is one full model+optimizer copy.

Optimizer state (FusedAdamW master/exp_avg/exp_avg_sq) shards exactly like
its parameter, so the same concat+slice applies positionally.
"""

import glob
import os
import re
from typing import Dict, List, Optional

import torch

from dlrover_amd.common.log import logger

SHARDING_KEY = "_sharding"  # {model_key: "shard0" | "replicated"}


def list_shard_files(path: str) -> List[str]:
    files = glob.glob(os.path.join(path, "rank_*.pt"))

    def rank_of(f):
        m = re.search(r"rank_(\d+)\.pt$", f)
        return int(m.group(1)) if m else 1 << 30

    return sorted(files, key=rank_of)


def _concat_model_states(shards: List[dict]) -> Dict[str, torch.Tensor]:
    """Merge old local shards into full tensors using the sharding tags."""
    tags = shards[0].get(SHARDING_KEY, {})
    full: Dict[str, torch.Tensor] = {}
    for key, first in shards[0]["model"].items():
        if tags.get(key, "replicated") == "shard0":
            pieces = [s["model"][key] for s in shards if s["model"][key].numel()]
            full[key] = torch.cat(pieces, dim=0)
        else:
            full[key] = first
    return full


def _concat_optim_states(shards: List[dict]) -> dict:
    """Merge FusedAdamW/torch optimizer state positionally (state tensors
    shard like their params on dim 0)."""
    base = shards[0].get("optimizer") or {}
    if not base:
        return {}
    out = {"state": {}, "param_groups": base.get("param_groups", [])}
    for idx in base.get("state", {}):
        merged = {}
        for k, v in base["state"][idx].items():
            if torch.is_tensor(v) and v.dim() >= 1:
                pieces = [
                    s["optimizer"]["state"][idx][k]
                    for s in shards
                    if idx in s["optimizer"]["state"]
                    and s["optimizer"]["state"][idx][k].numel()
                ]
                merged[k] = torch.cat(pieces, dim=0)
            else:
                merged[k] = v
        out["state"][idx] = merged
    return out


def _slice_for_rank(full: torch.Tensor, offsets: List[int], rank: int,
                    local_rows: int) -> torch.Tensor:
    start = offsets[rank]
    return full.narrow(0, start, local_rows)


def gather_new_offsets(local_rows: int, group=None) -> List[int]:
    """All-gather this world's dim-0 row counts -> start offsets per rank."""
    import torch.distributed as dist

    ws = dist.get_world_size(group)
    # an NCCL(-only) group cannot gather CPU tensors: stage on the device
    dev = "cpu"
    if "nccl" in str(dist.get_backend(group)).lower():
        dev = f"cuda:{torch.cuda.current_device()}"
    t = torch.tensor([local_rows], dtype=torch.long, device=dev)
    out = [torch.zeros(1, dtype=torch.long, device=dev) for _ in range(ws)]
    dist.all_gather(out, t, group=group)
    rows = [int(o.item()) for o in out]
    offsets, acc = [], 0
    for r in rows:
        offsets.append(acc)
        acc += r
    return offsets


def load_resharded(engine, model, optimizer, path: str) -> Optional[dict]:
    """Restore from a checkpoint written at a DIFFERENT world size.

    engine: ShardedCheckpointEngine (provides _named_tensors/load_into glue).
    """
    import torch.distributed as dist

    files = list_shard_files(path)
    if not files:
        return None
    logger.info("UCP reshard: %s old shards -> world %s",
                len(files), dist.get_world_size())
    shards = [torch.load(f, map_location="cpu", weights_only=False) for f in files]
    full_model = _concat_model_states(shards)
    full_optim = _concat_optim_states(shards)
    tags = shards[0].get(SHARDING_KEY, {})
    rank = dist.get_rank()

    from dlrover_amd.trainer.flash_checkpoint.engine import _to_local

    live = dict(engine._named_tensors(model))
    offsets_cache: Dict[int, List[int]] = {}
    with torch.no_grad():
        for name, saved in full_model.items():
            if name not in live:
                continue
            dst = _to_local(live[name].data)
            if tags.get(name, "replicated") == "shard0":
                # cache keyed by FULL dim-0 (rank-uniform): every rank takes
                # the same hit/miss sequence, so the all_gather stays
                # collective-consistent
                offs = offsets_cache.get(saved.shape[0])
                if offs is None:
                    offs = gather_new_offsets(dst.shape[0])
                    offsets_cache[saved.shape[0]] = offs
                my = _slice_for_rank(saved, offs, rank, dst.shape[0])
            else:
                my = saved
            dst.copy_(my.to(dst.device))

    if optimizer is not None and full_optim:
        # slice optimizer state to match the NEW local param shapes
        params = [p for g in optimizer.param_groups for p in g["params"]]
        new_state = {"state": {}, "param_groups": full_optim["param_groups"]}
        for g_new, g_old in zip(
            optimizer.state_dict()["param_groups"], full_optim["param_groups"]
        ):
            g_old["params"] = g_new["params"]
        for idx, st in full_optim["state"].items():
            if idx >= len(params):
                continue
            local = _to_local(params[idx])
            # the all_gather must be UNCONDITIONAL per state index: gating it
            # on v.shape[0] != local.shape[0] (a per-rank fact — one rank can
            # hold all rows while others hold 0) deadlocks the collective
            offs = gather_new_offsets(local.shape[0])
            sliced = {}
            for k, v in st.items():
                if torch.is_tensor(v) and v.dim() >= 1 and v.shape[0] != local.shape[0]:
                    sliced[k] = _slice_for_rank(v, offs, rank, local.shape[0]).clone()
                else:
                    sliced[k] = v
            new_state["state"][idx] = sliced
        optimizer.load_state_dict(new_state)

    # params omitted as _derived (bf16 twins of the fp32 master): rebuild
    # them from the freshly resharded optimizer state
    derived = shards[0].get("_derived") or {}
    if derived and optimizer is not None:
        params = [p for g in optimizer.param_groups for p in g["params"]]
        name2p = dict(model.named_parameters())
        with torch.no_grad():
            for name, idx in derived.items():
                if name in name2p and idx < len(params):
                    st = optimizer.state.get(params[idx], {})
                    if "master_param" in st:
                        dst = _to_local(name2p[name].data)
                        dst.copy_(st["master_param"].to(dst.dtype))

    meta = {k: v for k, v in shards[0].items()
            if not isinstance(v, dict) or k == "parallel"}
    meta["step"] = shards[0].get("step", 0)
    return meta
