"""Python stack tracing on hang (diagnosis/py_tracer.py): worker installs a
SIGUSR2 faulthandler dump; the agent signals, collects, and aggregates
identical stacks across ranks."""

import os
import subprocess
import sys
import textwrap
import time

from dlrover_amd.diagnosis import py_tracer

WORKER = textwrap.dedent(
    """
    import os, sys, time
    sys.path.insert(0, os.environ["REPO_ROOT"])
    import dlrover_amd  # auto-installs the tracer from DLROVER_PY_TRACER_DIR

    def stuck_in_allreduce():
        print("ready", flush=True)
        time.sleep(60)

    stuck_in_allreduce()
    """
)


def _spawn(rank, dump_dir):
    env = dict(os.environ)
    env.update(
        REPO_ROOT=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        DLROVER_PY_TRACER_DIR=dump_dir,
        RANK=str(rank),
    )
    return subprocess.Popen(
        [sys.executable, "-c", WORKER], env=env, stdout=subprocess.PIPE
    )


def test_dump_and_aggregate(tmp_path):
    dump_dir = str(tmp_path)
    procs = {lr: _spawn(lr, dump_dir) for lr in range(2)}
    try:
        for p in procs.values():
            assert p.stdout.readline().strip() == b"ready"
        pids = {lr: p.pid for lr, p in procs.items()}
        # under parallel test load a signal can land mid-call and produce
        # transiently different stacks — retry until both ranks coalesce
        agg = ""
        for _ in range(3):
            stacks = py_tracer.dump_worker_stacks(pids, dump_dir, timeout=15.0)
            assert set(stacks) == {0, 1}
            for text in stacks.values():
                assert "stuck_in_allreduce" in text, text
            agg = py_tracer.aggregate_stacks(stacks)
            if "ranks 0-1 (2 rank(s))" in agg:
                break
            time.sleep(0.5)
        # both ranks share the stack -> one group headed "ranks 0-1"
        assert "ranks 0-1 (2 rank(s))" in agg, agg
        assert agg.count("stuck_in_allreduce") == 1, agg
    finally:
        for p in procs.values():
            p.kill()
            p.wait()


def test_fmt_ranks():
    assert py_tracer._fmt_ranks([0, 1, 2, 5]) == "0-2,5"
    assert py_tracer._fmt_ranks([3]) == "3"


def test_aggregate_groups_distinct_stacks():
    a = 'Current thread 0x1 (most recent call first):\n  File "x.py", line 1 in aa\n'
    b = 'Current thread 0x2 (most recent call first):\n  File "y.py", line 9 in bb\n'
    agg = py_tracer.aggregate_stacks({0: a, 1: b, 2: a})
    assert "ranks 0,2 (2 rank(s))" in agg
    assert "ranks 1 (1 rank(s))" in agg
