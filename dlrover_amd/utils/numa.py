"""NUMA affinity for worker processes (ref: dlrover/python/util/numa_util.py
+ scripts/dlrover_run_affinity.sh; flag --numa-affinity).

On an MI355X node each GPU hangs off a specific NUMA domain; binding the
worker's CPU threads (dataloader, RCCL proxy, D2H drain) to the GPU's local
node avoids cross-socket traffic on the PCIe/host path the flash-checkpoint
drain depends on.
"""

import glob
import os
from typing import Dict, List, Optional

from dlrover_amd.common.log import logger


def gpu_numa_node(card_index: int) -> Optional[int]:
    """NUMA node of an amdgpu card via sysfs."""
    for pattern in (
        f"/sys/class/drm/card{card_index}/device/numa_node",
        f"/sys/class/kfd/kfd/topology/nodes/{card_index + 1}/properties",
    ):
        for path in glob.glob(pattern):
            try:
                if path.endswith("numa_node"):
                    node = int(open(path).read().strip())
                    return node if node >= 0 else None
            except (OSError, ValueError):
                continue
    return None


def numa_cpus(node: int) -> List[int]:
    try:
        text = open(f"/sys/devices/system/node/node{node}/cpulist").read().strip()
    except OSError:
        return []
    cpus: List[int] = []
    for part in text.split(","):
        if "-" in part:
            lo, hi = part.split("-")
            cpus.extend(range(int(lo), int(hi) + 1))
        elif part:
            cpus.append(int(part))
    return cpus


def bind_to_gpu_numa(local_rank: int) -> bool:
    """Pin this process to the NUMA node local to its GPU. Returns True if a
    binding was applied."""
    node = gpu_numa_node(local_rank)
    if node is None:
        return False
    cpus = numa_cpus(node)
    if not cpus:
        return False
    try:
        os.sched_setaffinity(0, cpus)
        logger.info(
            "numa affinity: local_rank %s -> node %s (%s cpus)",
            local_rank, node, len(cpus),
        )
        return True
    except OSError:
        return False


def worker_affinity_env(nproc: int) -> Dict[int, Dict[str, str]]:
    """Per-local-rank env asking the worker to self-bind at startup."""
    return {
        r: {"DLROVER_NUMA_BIND": "1", "DLROVER_NUMA_LOCAL_RANK": str(r)}
        for r in range(nproc)
    }


def maybe_bind_from_env():
    """Called from worker entry (models/trainer import time)."""
    if os.getenv("DLROVER_NUMA_BIND") == "1":
        bind_to_gpu_numa(int(os.getenv("DLROVER_NUMA_LOCAL_RANK",
                                       os.getenv("LOCAL_RANK", "0"))))
