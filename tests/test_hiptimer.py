"""hiptimer: metrics parsing + collector logic (CPU) and LD_PRELOAD
interception on a real GPU workload (gpu-marked)."""

import json
import os
import subprocess
import sys
import time

import pytest

from dlrover_amd import xpu_timer


def _write_prom(path, hang, count=10):
    with open(path, "w") as f:
        f.write(f"XPU_TIMER_COMMON_HANG {hang}\n")
        f.write("hiptimer_hang_since_seconds 123.0\n")
        f.write(f'hiptimer_op_count{{cat="kernel"}} {count}\n')
        f.write('hiptimer_op_ms_total{cat="kernel"} 42.5\n')


def test_parse_metrics(tmp_path):
    p = tmp_path / "hiptimer_0.prom"
    _write_prom(p, 0)
    m = xpu_timer.parse_metrics_file(str(p))
    assert m["XPU_TIMER_COMMON_HANG"] == 0
    assert m['hiptimer_op_count{cat="kernel"}'] == 10
    assert m['hiptimer_op_ms_total{cat="kernel"}'] == 42.5


def test_collector_hang_requires_all_ranks(tmp_path):
    _write_prom(tmp_path / "hiptimer_0.prom", 1)
    _write_prom(tmp_path / "hiptimer_1.prom", 0)
    col = xpu_timer.HiptimerCollector(str(tmp_path))
    assert col.node_hang_state()["hang"] is False
    _write_prom(tmp_path / "hiptimer_1.prom", 1)
    state = col.node_hang_state()
    assert state["hang"] is True and state["ranks"] == 2


def test_collector_empty_dir(tmp_path):
    col = xpu_timer.HiptimerCollector(str(tmp_path))
    assert col.node_hang_state()["hang"] is False


@pytest.mark.gpu
def test_hiptimer_intercepts_real_kernels(tmp_path):
    """LD_PRELOAD into a small torch-GPU workload; metrics must show kernel
    and memcpy activity (the 'native code is loaded' check for hiptimer)."""
    assert xpu_timer.available(), "libhiptimer.so not built"
    metrics_dir = str(tmp_path / "m")
    env = xpu_timer.preload_env(metrics_dir, base_env=dict(os.environ))
    env["HIPTIMER_DUMP_INTERVAL"] = "1"
    env["RANK"] = "0"
    code = (
        "import torch;"
        "a = torch.randn(2048, 2048, device='cuda', dtype=torch.bfloat16);"
        "b = a @ a;"
        "c = (a + b).sum();"
        "torch.cuda.synchronize();"
        "import time; time.sleep(2.5);"
        "print('done', c.item())"
    )
    out = subprocess.run(
        [sys.executable, "-c", code], env=env, capture_output=True, text=True,
        timeout=180,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    m = xpu_timer.parse_metrics_file(os.path.join(metrics_dir, "hiptimer_0.prom"))
    assert m, "no metrics dumped"
    assert m.get("hiptimer_launched_total", 0) > 0
    assert m.get('hiptimer_op_count{cat="kernel"}', 0) > 0
    assert m.get("XPU_TIMER_COMMON_HANG") == 0


def test_rccl_env_comm_summary():
    from dlrover_amd.xpu_timer import rccl_env

    metrics = {
        'hiptimer_comm_calls{comm="0xa",nranks="8",rank="3",alive="1"}': 120.0,
        'hiptimer_comm_elems{comm="0xa",nranks="8",rank="3",alive="1"}': 4.0e9,
        'hiptimer_comm_calls{comm="0xb",nranks="2",rank="1",alive="0"}': 7.0,
        'hiptimer_comm_elems{comm="0xb",nranks="2",rank="1",alive="0"}': 1.0e6,
        "hiptimer_launched_total": 99.0,
    }
    comms = rccl_env.summarize_comms(metrics)
    assert [c["comm"] for c in comms] == ["0xa", "0xb"]  # traffic-sorted
    assert comms[0]["nranks"] == 8 and comms[0]["rank"] == 3 and comms[0]["alive"]
    assert comms[1]["calls"] == 7.0 and not comms[1]["alive"]
    report = rccl_env.format_comm_report(metrics)
    assert "nranks=8" in report and "destroyed" in report
    assert rccl_env.format_comm_report({"x": 1.0}) is None


def test_rccl_env_bandwidth_expectations(monkeypatch):
    from dlrover_amd.xpu_timer import rccl_env

    assert rccl_env.expected_ring_busbw_gbps() == 153.0
    # all-reduce algbw bound: busbw * n / (2(n-1)) -> 8/14 of link bw at n=8
    assert abs(rccl_env.expected_allreduce_algbw_gbps(8) - 153.0 * 8 / 14) < 1e-9
    monkeypatch.setenv("NCCL_MIN_NCHANNELS", "4")
    env = rccl_env.effective_env()
    assert env["NCCL_MIN_NCHANNELS"] == "4"


def test_prometheus_exporter_serves_aggregated_metrics(tmp_path):
    """HTTP scrape endpoint (ref daemon :18889): per-rank .prom files are
    merged with a rank label in standard exposition format."""
    import urllib.request

    from dlrover_amd import xpu_timer

    (tmp_path / "hiptimer_0.prom").write_text(
        "XPU_TIMER_COMMON_HANG 0\nhiptimer_gemm_tflops 123.45\n"
        'hiptimer_op_count{cat="gemm"} 7\n'
    )
    (tmp_path / "hiptimer_1.prom").write_text("XPU_TIMER_COMMON_HANG 1\n")
    exp = xpu_timer.PrometheusExporter(str(tmp_path), port=0,
                                       host="127.0.0.1").start()
    try:
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{exp.port}/metrics", timeout=10
        ).read().decode()
        assert 'XPU_TIMER_COMMON_HANG{rank="0"} 0' in body
        assert 'XPU_TIMER_COMMON_HANG{rank="1"} 1' in body
        assert 'hiptimer_gemm_tflops{rank="0"} 123.45' in body
        assert 'hiptimer_op_count{rank="0",cat="gemm"} 7' in body
    finally:
        exp.stop()


def test_crash_backtrace_dump(tmp_path):
    """Fatal signals leave a native backtrace in the metrics dir before the
    default handler runs (ref signal_handler.cc)."""
    import os
    import signal
    import subprocess
    import sys

    from dlrover_amd import xpu_timer

    lib = xpu_timer.lib_path() if hasattr(xpu_timer, "lib_path") else None
    if lib is None:
        import dlrover_amd

        lib = os.path.join(os.path.dirname(dlrover_amd.__file__),
                           "xpu_timer", "libhiptimer.so")
    env = dict(os.environ)
    env.update({
        "LD_PRELOAD": lib,
        "HIPTIMER_METRICS_DIR": str(tmp_path),
        "HIPTIMER_NO_POLLER": "1",
        "RANK": "0",
    })
    proc = subprocess.run(
        [sys.executable, "-c",
         "import ctypes; lib = ctypes.CDLL(None); lib.hipFree(None); "
         "ctypes.string_at(0)"],  # hipFree = interposed (installs handler)
        env=env, capture_output=True, timeout=120,
    )
    assert proc.returncode != 0
    crash = tmp_path / "crash_0.txt"
    assert crash.exists(), list(tmp_path.iterdir())
    txt = crash.read_text()
    assert "signal 11" in txt and "backtrace" in txt
    assert proc.returncode in (-signal.SIGSEGV, 139, -11, 1), proc.returncode
