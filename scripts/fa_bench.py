"""Time FA forward variants at the flagship shape (B1 H32 HKV8 S4096 D128).

TF = 2 * B*H*S*(S+1)/2 * D * 2 ops (QK^T + PV, causal half) / time.
Run on a GPU box: python3 scripts/fa_bench.py [--bwd]
"""

import argparse
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from dlrover_amd.ops.api import hip_ops  # noqa: E402


def flops_causal(B, H, S, D):
    return 2 * 2 * B * H * (S * (S + 1) / 2) * D


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--B", type=int, default=1)
    p.add_argument("--H", type=int, default=32)
    p.add_argument("--HKV", type=int, default=8)
    p.add_argument("--S", type=int, default=4096)
    p.add_argument("--bwd", action="store_true")
    args = p.parse_args()
    B, H, HKV, S, D = args.B, args.H, args.HKV, args.S, 128
    torch.manual_seed(0)
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, HKV, S, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, HKV, S, D, device="cuda", dtype=torch.bfloat16)
    scale = 1 / math.sqrt(D)
    fl = flops_causal(B, H, S, D)
    ext = hip_ops()

    # numerics cross-check v3 vs v1 at this exact shape first
    os.environ.pop("DLROVER_FA_V2", None)
    out3, lse3 = ext.flash_attn_fwd(q, k, v, scale)

    results = {}
    t = bench(lambda: ext.flash_attn_fwd(q, k, v, scale))
    results["default(v3)" if S % 256 == 0 else "default(v1)"] = t
    if args.bwd:
        dout = torch.randn_like(out3)
        t = bench(
            lambda: ext.flash_attn_bwd(q, k, v, out3.contiguous(), dout,
                                       lse3, scale),
            iters=10, warmup=3,
        )
        results["bwd(dq+dkv)"] = t
    for name, t in results.items():
        print(f"{name}: {t * 1e3:.3f} ms  {fl / t / 1e12:.1f} TF")


if __name__ == "__main__":
    main()
