"""Coverage for the offline analysis tools (timeline/GEMM/stack utilities
mirroring the reference's py_xpu_timer toolbox)."""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, **kw):
    return subprocess.run([sys.executable] + args, capture_output=True,
                          text=True, cwd=ROOT, **kw)


def test_gemm_report_parses_prom(tmp_path):
    prom = tmp_path / "hiptimer_0.prom"
    prom.write_text(
        'hiptimer_kernel_count{name="gemm_m4096_n4096_k4096_b1"} 10\n'
        'hiptimer_kernel_ms_total{name="gemm_m4096_n4096_k4096_b1"} 100.0\n'
        'hiptimer_kernel_count{name="other_kernel"} 5\n'
    )
    out = _run(["tools/gemm_report.py", str(prom)])
    assert out.returncode == 0, out.stderr
    # 2*4096^3*10 flops / 0.1 s = 13.7 TFLOPS
    line = [ln for ln in out.stdout.splitlines() if "4096" in ln][0]
    assert "13.7" in line, out.stdout


def test_stack_collapse_folds_faulthandler_dumps(tmp_path):
    dump = tmp_path / "stacks_0.txt"
    dump.write_text(
        "Current thread 0x00007f (most recent call first):\n"
        '  File "/a/b/inner.py", line 10 in inner_fn\n'
        '  File "/a/b/outer.py", line 20 in outer_fn\n'
    )
    out = _run(["tools/stack_collapse.py", str(dump)])
    assert out.returncode == 0, out.stderr
    line = out.stdout.strip()
    # root-first ordering, counted once
    assert line.endswith(" 1")
    assert line.index("outer_fn") < line.index("inner_fn")


def test_dump_timeline_cli_collects(tmp_path):
    mdir = tmp_path / "hiptimer_job"
    mdir.mkdir()
    import threading
    import time

    def fake_rank():
        # a preloaded rank notices the mtime edge and dumps
        flag = mdir / "dump_timeline_all"
        deadline = time.time() + 10
        while time.time() < deadline and not flag.exists():
            time.sleep(0.05)
        (mdir / "timeline_0.json").write_text(json.dumps({"traceEvents": []}))

    t = threading.Thread(target=fake_rank)
    t.start()
    out = _run(["tools/dump_timeline.py", "--metrics-dir", str(mdir),
                "--timeout", "15"])
    t.join()
    assert out.returncode == 0, out.stderr
    assert "timeline_0.json" in out.stdout
