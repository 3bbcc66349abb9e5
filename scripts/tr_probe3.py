"""Decode tr_b16 probe r3: which (4x4 tile, row, col) each lane receives
when tile-base address bits VARY inside a 16-lane group.

LDS word i holds raw bits i, so every returned short identifies its source:
  tile = v // 16, row = (v % 16) // 4, col = v % 4   (32-byte packed tiles)
Reads:
  A: addr = lane*32            (tile base varies per LANE, col bits 0)
  B: addr = (lane>>2)*32 + (lane&3)*2   (tile per QUAD, col per lane)
  C: addr = (lane>>4)*512 + (lane&3)*2, offset:64    (tile per GROUP+imm)
"""

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: F401

from dlrover_amd.ops.api import hip_ops


def decode(v):
    v = int(v) & 0xFFFF
    return (v // 16, (v % 16) // 4, v % 4)


def main():
    out = hip_ops().tr_b16_probe3().cpu()
    for name, base in (("A", 0), ("B", 4), ("C", 8)):
        print(f"--- read {name} ---")
        for lane in range(64):
            vals = [decode(out[lane, base + j]) for j in range(4)]
            print(f"lane {lane:2d}: " + " ".join(
                f"t{t:3d} r{r} c{c}" for t, r, c in vals))


if __name__ == "__main__":
    main()
