#!/usr/bin/env python3
"""Trigger and collect hiptimer kernel-trace timeline dumps.

Counterpart of the reference's `xpu_timer_dump_timeline` (py_xpu_timer
dump_timeline.py:70): touches the shared mtime-edge flag every preloaded
local rank watches, waits for the chrome-trace JSON files to appear, and
prints their paths (load them in perfetto / chrome://tracing).

Usage:
    python tools/dump_timeline.py [--metrics-dir /tmp/hiptimer_<job>]
                                  [--timeout 30]
For remote/multi-host jobs, run it on each host, or push the master-side
DUMP_TIMELINE diagnosis action which fans out the same flag per node.
"""

import argparse
import glob
import os
import sys
import time


def find_metrics_dir() -> str:
    env = os.getenv("HIPTIMER_METRICS_DIR", "")
    if env:
        return env
    job = os.getenv("ELASTIC_JOB_NAME", "")
    if job and os.path.isdir(f"/tmp/hiptimer_{job}"):
        return f"/tmp/hiptimer_{job}"
    candidates = sorted(glob.glob("/tmp/hiptimer_*"), key=os.path.getmtime)
    if candidates:
        return candidates[-1]
    return "/tmp/hiptimer"


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--metrics-dir", default="", help="hiptimer metrics dir")
    p.add_argument("--timeout", type=float, default=30.0)
    args = p.parse_args()

    mdir = args.metrics_dir or find_metrics_dir()
    if not os.path.isdir(mdir):
        print(f"no hiptimer metrics dir at {mdir} — is a job running with "
              "--hiptimer?", file=sys.stderr)
        return 1
    t0 = time.time()
    # mtime-edge-triggered shared flag: every preloaded rank dumps once
    flag = os.path.join(mdir, "dump_timeline_all")
    with open(flag, "a"):
        os.utime(flag, None)
    print(f"touched {flag}; waiting for dumps...", file=sys.stderr)

    deadline = time.time() + args.timeout
    seen = set()
    while time.time() < deadline:
        for path in glob.glob(os.path.join(mdir, "timeline_*.json")):
            if path not in seen and os.path.getmtime(path) >= t0 - 1:
                seen.add(path)
                print(path)
        if seen:
            # ranks dump within one 20 ms poll of each other; linger briefly
            time.sleep(1.0)
            for path in glob.glob(os.path.join(mdir, "timeline_*.json")):
                if path not in seen and os.path.getmtime(path) >= t0 - 1:
                    seen.add(path)
                    print(path)
            return 0
        time.sleep(0.5)
    print("no timeline dumped (are workers preloaded with libhiptimer and "
          "actively launching kernels?)", file=sys.stderr)
    return 2


if __name__ == "__main__":
    sys.exit(main())
