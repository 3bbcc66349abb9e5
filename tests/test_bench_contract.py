"""bench.py driver contract: one JSON line on stdout with the agreed keys,
works standalone and under torch.distributed.run."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _parse_bench(stdout: str) -> dict:
    lines = [l for l in stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected exactly ONE json line, got {len(lines)}"
    return json.loads(lines[0])


@pytest.mark.timeout(420)
def test_bench_default_contract(tmp_path):
    env = dict(os.environ, MASTER_PORT="29741",
               DLROVER_IPC_SOCKET_DIR=str(tmp_path / "ipc"),
               ELASTIC_JOB_NAME=f"bench{os.getpid()}")
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "tiny", "--steps", "4",
         "--warmup", "1", "--seq", "64", "--ckpt-interval", "2"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    r = _parse_bench(out.stdout)
    assert REQUIRED_KEYS <= set(r)
    assert r["n_gpus"] == 1 and r["steps"] == 4 and r["warmup"] == 1
    assert r["higher_is_better"] is True and r["scaling"] == "weak"
    assert r["data"] == "synthetic"
    assert 0 < r["value"] <= 100  # goodput percent
    assert r["vs_baseline"] == pytest.approx(r["value"] / 95.0, rel=1e-3)
    cfg = r["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism",
                "tokens_per_s", "ckpt_save_blocking_s", "ckpt_restore_s"):
        assert key in cfg, key


@pytest.mark.timeout(420)
def test_bench_under_torchrun(tmp_path):
    env = dict(os.environ, DLROVER_IPC_SOCKET_DIR=str(tmp_path / "ipc"),
               ELASTIC_JOB_NAME=f"bench2{os.getpid()}")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29743", "bench.py", "--gpus", "2", "--model",
         "tiny", "--steps", "4", "--warmup", "1", "--seq", "64",
         "--ckpt-interval", "2"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=360,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    r = _parse_bench(out.stdout)
    assert r["n_gpus"] == 2
    assert r["config"]["global_batch"] == 2 * 2  # ws=2 x default batch 2 (weak scaling)


@pytest.mark.timeout(120)
def test_bench_refuses_world_size_mismatch():
    """--gpus N with a conflicting torchrun WORLD_SIZE must refuse rather
    than silently measure a different world size."""
    env = dict(os.environ, RANK="0", WORLD_SIZE="1", LOCAL_RANK="0",
               MASTER_ADDR="127.0.0.1", MASTER_PORT="29742")
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "2", "--model", "tiny",
         "--steps", "2", "--warmup", "0", "--seq", "16"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=90,
    )
    assert out.returncode != 0
    assert "refusing" in out.stderr or "WORLD_SIZE" in out.stderr, out.stderr[-1500:]
