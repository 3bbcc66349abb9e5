#!/usr/bin/env python3
"""Per-shape GEMM report from hiptimer metrics (analysis counterpart of the
reference's py_xpu_timer parse_matmul): reads one or more hiptimer_*.prom
files and prints a TFLOPS table per (m, n, k, batch) shape.

Usage: python tools/gemm_report.py /tmp/hiptimer_<job>/hiptimer_0.prom ...
"""

import re
import sys

_SHAPE = re.compile(
    r'hiptimer_kernel_(count|ms_total)\{name="gemm_m(\d+)_n(\d+)_k(\d+)_b(\d+)"\} '
    r"([0-9.eE+-]+)"
)


def parse(paths):
    shapes = {}
    for path in paths:
        try:
            text = open(path).read()
        except OSError as e:
            print(f"skip {path}: {e}", file=sys.stderr)
            continue
        for kind, m, n, k, b, val in _SHAPE.findall(text):
            key = (int(m), int(n), int(k), int(b))
            entry = shapes.setdefault(key, {"count": 0.0, "ms": 0.0})
            entry["count" if kind == "count" else "ms"] += float(val)
    return shapes


def main():
    if len(sys.argv) < 2:
        raise SystemExit(__doc__)
    shapes = parse(sys.argv[1:])
    if not shapes:
        print("no per-shape GEMM entries found (fp32 models may route via "
              "rocBLAS, which has no hipblasLt layout to read)")
        return
    rows = []
    for (m, n, k, b), e in shapes.items():
        flops = 2.0 * m * n * k * b * e["count"]
        tf = flops / (e["ms"] / 1e3) / 1e12 if e["ms"] > 0 else 0.0
        rows.append((e["ms"], m, n, k, b, int(e["count"]), tf))
    rows.sort(reverse=True)
    print(f"{'ms_total':>10} {'m':>7} {'n':>7} {'k':>7} {'batch':>5} "
          f"{'calls':>7} {'TFLOPS':>8}")
    for ms, m, n, k, b, cnt, tf in rows:
        print(f"{ms:10.2f} {m:7d} {n:7d} {k:7d} {b:5d} {cnt:7d} {tf:8.1f}")


if __name__ == "__main__":
    main()
