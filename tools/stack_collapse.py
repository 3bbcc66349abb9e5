#!/usr/bin/env python3
"""Collapse py_tracer faulthandler dumps into flamegraph 'folded' format.

Counterpart of the reference's stack viewer (py_xpu_timer stack_viewer.py):
reads one or more SIGUSR2 stack dump files written by
dlrover_amd.diagnosis.py_tracer (faulthandler text), and emits one
``frame;frame;...;frame count`` line per distinct stack across all threads
and files — pipe into flamegraph.pl or load into speedscope.

Usage:
    python tools/stack_collapse.py /tmp/dlrover_pystacks/*.txt > out.folded
"""

import re
import sys

_THREAD_RE = re.compile(r"^(Current thread|Thread) 0x[0-9a-f]+")
_FRAME_RE = re.compile(r'File "(?P<file>[^"]+)", line (?P<line>\d+) in (?P<fn>.+)')


def collapse(texts):
    counts = {}
    for text in texts:
        cur = None
        for line in text.splitlines():
            s = line.strip()
            if _THREAD_RE.match(s):
                if cur:
                    key = ";".join(reversed(cur))  # root-first for flamegraphs
                    counts[key] = counts.get(key, 0) + 1
                cur = []
            elif cur is not None:
                m = _FRAME_RE.match(s)
                if m:
                    short = m.group("file").rsplit("/", 1)[-1]
                    cur.append(f"{m.group('fn')} ({short}:{m.group('line')})")
        if cur:
            key = ";".join(reversed(cur))
            counts[key] = counts.get(key, 0) + 1
    return counts


def main() -> int:
    paths = sys.argv[1:]
    if not paths:
        print(__doc__, file=sys.stderr)
        return 1
    texts = []
    for p in paths:
        try:
            with open(p, errors="replace") as f:
                texts.append(f.read())
        except OSError as e:
            print(f"skip {p}: {e}", file=sys.stderr)
    for key, n in sorted(collapse(texts).items(), key=lambda kv: -kv[1]):
        print(f"{key} {n}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
