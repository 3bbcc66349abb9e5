"""hiptimer launch wrapper (ref: xpu_timer/py_xpu_timer/bin/xpu_timer_launch
— set LD_PRELOAD + config env, exec the training command).

Usage: python -m dlrover_amd.xpu_timer.launch [--metrics-dir D]
       [--hang-secs S] -- <command> [args...]
"""

import argparse
import os
import sys

from dlrover_amd import xpu_timer


def main(argv=None) -> int:
    argv = list(sys.argv[1:] if argv is None else argv)
    if "--" in argv:
        split = argv.index("--")
        own, cmd = argv[:split], argv[split + 1 :]
    else:
        own, cmd = [], argv
    p = argparse.ArgumentParser("hiptimer-launch")
    p.add_argument("--metrics-dir", default="/tmp/hiptimer")
    p.add_argument("--hang-secs", type=float, default=60.0)
    p.add_argument("--dump-interval", type=float, default=5.0)
    args = p.parse_args(own)
    if not cmd:
        p.error("no command given (use: ... -- python train.py)")
    if not xpu_timer.available():
        print("libhiptimer.so is not built (run __graft_entry__.build())",
              file=sys.stderr)
        return 2
    env = xpu_timer.preload_env(
        args.metrics_dir, hang_secs=args.hang_secs, base_env=dict(os.environ)
    )
    env["HIPTIMER_DUMP_INTERVAL"] = str(args.dump_interval)
    os.execvpe(cmd[0], cmd, env)


if __name__ == "__main__":
    sys.exit(main())
