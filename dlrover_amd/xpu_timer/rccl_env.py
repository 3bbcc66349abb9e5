"""RCCL communicator / tuning introspection.

MI355X-native counterpart of the reference's NCCL param parser
(xpu_timer's nccl introspection feeds the diagnostician which process group
a hang sits in, and what tuning the job ran with). Two sources:

- libhiptimer interposes ncclCommInitRank[Config]/ncclCommDestroy and every
  collective, exporting per-communicator (nranks, rank, calls, elems) as
  ``hiptimer_comm_*`` metrics — :func:`summarize_comms` parses those back
  into structured records so logs can say "the 8-rank DP comm stalled, the
  2-rank TP comm kept moving".
- :func:`effective_env` snapshots the RCCL/HSA tuning environment that
  actually applied to the job (the knobs that matter on xGMI, not the
  NVSwitch ones).

Ring bandwidth expectations come from GpuConstant's xGMI facts: a ring
collective is bound by one link per direction, so busbw ≈ 153 GB/s/link and
algorithm bandwidth = busbw * n/(2(n-1)) for all-reduce.
"""

import os
import re
from typing import Dict, List, Optional

from dlrover_amd.common.constants import GpuConstant

# the tuning surface that matters for RCCL over xGMI (single MI355X node)
RCCL_TUNING_VARS = [
    "NCCL_DEBUG",
    "NCCL_ALGO",
    "NCCL_PROTO",
    "NCCL_MIN_NCHANNELS",
    "NCCL_MAX_NCHANNELS",
    "NCCL_P2P_LEVEL",
    "NCCL_P2P_DISABLE",
    "NCCL_SHM_DISABLE",
    "NCCL_IB_DISABLE",
    "NCCL_SOCKET_IFNAME",
    "NCCL_LAUNCH_MODE",
    "RCCL_MSCCL_ENABLE",
    "RCCL_MSCCLPP_ENABLE",
    "HSA_ENABLE_IPC_MODE_LEGACY",  # dmabuf IPC: required 0 on this driver
    "HSA_FORCE_FINE_GRAIN_PCIE",
]


def effective_env() -> Dict[str, str]:
    """The RCCL/HSA tuning vars that are actually set (unset ones omitted —
    RCCL's in-library defaults apply)."""
    return {k: os.environ[k] for k in RCCL_TUNING_VARS if k in os.environ}


def expected_ring_busbw_gbps() -> float:
    """Per-link bound for a ring collective on xGMI (one link each way)."""
    return GpuConstant.XGMI_LINK_GBPS


def expected_allreduce_algbw_gbps(nranks: int) -> float:
    """Ring all-reduce algorithm bandwidth bound: busbw * n / (2 (n-1))."""
    if nranks <= 1:
        return float("inf")
    return GpuConstant.XGMI_LINK_GBPS * nranks / (2.0 * (nranks - 1))


_COMM_RE = re.compile(
    r'hiptimer_comm_(calls|elems)\{comm="(?P<comm>[^"]+)",nranks="(?P<nranks>\d+)"'
    r',rank="(?P<rank>\d+)",alive="(?P<alive>\d)"\}'
)


def summarize_comms(metrics: Dict[str, float]) -> List[dict]:
    """Turn parse_metrics_file() output into one record per communicator:
    {comm, nranks, rank, alive, calls, elems}, largest-traffic first."""
    comms: Dict[str, dict] = {}
    for key, val in metrics.items():
        m = _COMM_RE.match(key)
        if not m:
            continue
        c = comms.setdefault(
            m.group("comm"),
            {
                "comm": m.group("comm"),
                "nranks": int(m.group("nranks")),
                "rank": int(m.group("rank")),
                "alive": m.group("alive") == "1",
                "calls": 0.0,
                "elems": 0.0,
            },
        )
        c["calls" if key.startswith("hiptimer_comm_calls") else "elems"] = val
    return sorted(comms.values(), key=lambda c: -c["elems"])


def format_comm_report(metrics: Dict[str, float]) -> Optional[str]:
    """Human-oriented one-liner per comm for hang dossiers, or None if the
    preload recorded no communicators."""
    comms = summarize_comms(metrics)
    if not comms:
        return None
    lines = [
        f"comm {c['comm']} nranks={c['nranks']} rank={c['rank']} "
        f"calls={c['calls']:.0f} elems={c['elems']:.3g} "
        f"{'alive' if c['alive'] else 'destroyed'}"
        for c in comms
    ]
    env = effective_env()
    if env:
        lines.append("tuning: " + " ".join(f"{k}={v}" for k, v in sorted(env.items())))
    return "\n".join(lines)
