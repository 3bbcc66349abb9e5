#!/usr/bin/env python3
"""Replay recorded GEMM shapes standalone and measure achievable TFLOPS.

Counterpart of the reference's matmul replay (py_xpu_timer parse_matmul):
takes shapes either from hiptimer .prom files (gemm_m<m>_n<n>_k<k>_b<b>
labels) or from the command line, runs them with torch on cuda:0 (routes
through hipBLASLt on ROCm), and prints measured vs in-job TFLOPS — the gap
shows whether a slow GEMM was the shape itself or interference in the job.

Usage (on a GPU box):
    python tools/gemm_replay.py /tmp/hiptimer_<job>/hiptimer_0.prom
    python tools/gemm_replay.py --shape 4096x4096x4096 --dtype bf16
"""

import argparse
import sys

from gemm_report import parse  # same directory


def replay(m: int, n: int, k: int, b: int, dtype, iters: int = 50) -> float:
    import torch

    a = torch.randn(b, m, k, device="cuda", dtype=dtype)
    w = torch.randn(b, k, n, device="cuda", dtype=dtype)
    for _ in range(5):
        torch.bmm(a, w)
    torch.cuda.synchronize()
    import time

    t0 = time.perf_counter()
    for _ in range(iters):
        torch.bmm(a, w)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return 2.0 * m * n * k * b / dt / 1e12


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("prom_files", nargs="*", help="hiptimer .prom files")
    p.add_argument("--shape", action="append", default=[],
                   help="MxNxK[xB] to replay directly")
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp16", "fp32"])
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--top", type=int, default=10,
                   help="replay the top-N shapes by total ms")
    args = p.parse_args()

    import torch

    if not torch.cuda.is_available():
        print("no GPU visible — run on an MI355X box", file=sys.stderr)
        return 1
    dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
             "fp32": torch.float32}[args.dtype]

    jobs = []  # (m, n, k, b, in_job_tflops or None)
    for s in args.shape:
        dims = [int(x) for x in s.lower().split("x")]
        m, n, k = dims[:3]
        b = dims[3] if len(dims) > 3 else 1
        jobs.append((m, n, k, b, None))
    if args.prom_files:
        shapes = parse(args.prom_files)
        ranked = sorted(shapes.items(), key=lambda kv: -kv[1]["ms"])
        for (m, n, k, b), e in ranked[: args.top]:
            flops = 2.0 * m * n * k * b * e["count"]
            tf = flops / (e["ms"] / 1e3) / 1e12 if e["ms"] > 0 else None
            jobs.append((m, n, k, b, tf))
    if not jobs:
        raise SystemExit(__doc__)

    print(f"{'m':>7} {'n':>7} {'k':>7} {'b':>4} {'in-job TF':>10} "
          f"{'replay TF':>10} {'ratio':>6}")
    for m, n, k, b, in_job in jobs:
        got = replay(m, n, k, b, dtype, args.iters)
        ratio = f"{in_job / got:6.2f}" if in_job else "   n/a"
        ij = f"{in_job:10.1f}" if in_job else "       n/a"
        print(f"{m:7d} {n:7d} {k:7d} {b:4d} {ij} {got:10.1f} {ratio}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
