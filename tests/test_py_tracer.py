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


def test_py_runtime_tracer_gc_and_dataloader(tmp_path):
    """In-process GC pause + dataloader wait tracing (ref py_tracing_manager
    ring counters) exported as Prometheus text."""
    import gc
    import time

    from dlrover_amd.diagnosis.py_runtime_tracer import PyRuntimeTracer

    tr = PyRuntimeTracer(metrics_dir=str(tmp_path), interval=600)
    tr.start()
    try:
        for _ in range(3):
            gc.collect()

        def slow_batches():
            for i in range(4):
                time.sleep(0.01)
                yield i

        wrapped = tr.wrap_loader(slow_batches())
        assert list(wrapped) == [0, 1, 2, 3]
        m = tr.metrics()
        assert m['py_gc_collections{gen="2"}'] >= 3
        assert m["py_dataloader_batches"] == 4
        assert m["py_dataloader_wait_ms"] >= 30
    finally:
        tr.stop()
    prom = (tmp_path / "pymetrics_0.prom").read_text()
    assert "py_dataloader_batches 4" in prom


def test_py_metrics_served_by_prometheus_exporter(tmp_path):
    import urllib.request

    from dlrover_amd import xpu_timer
    from dlrover_amd.diagnosis.py_runtime_tracer import PyRuntimeTracer

    tr = PyRuntimeTracer(metrics_dir=str(tmp_path), interval=600)
    tr.dump()
    exp = xpu_timer.PrometheusExporter(str(tmp_path), port=0,
                                       host="127.0.0.1").start()
    try:
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{exp.port}/metrics", timeout=10
        ).read().decode()
        assert 'py_dataloader_batches{rank="0"}' in body
    finally:
        exp.stop()


def test_read_kernel_stacks_self():
    import os

    from dlrover_amd.diagnosis.py_tracer import read_kernel_stacks

    out = read_kernel_stacks({0: os.getpid()})
    # root in this container: our own kernel stacks are readable
    assert 0 in out and out[0]
