"""End-to-end: dlrover-run standalone, nanoGPT DDP world_size=2 on CPU/gloo,
with flash checkpoint and SIGKILL-restart recovery (BASELINE.json config #1).

Spawns the real CLI as a subprocess: local master subprocess + elastic agent
+ 2 gloo workers, all production code paths.
"""

import json
import os
import subprocess
import sys
import time
import uuid

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_cli(tmp_path, extra_env=None, steps=12, ckpt_interval=3, max_restarts=2,
             timeout=420, nproc=2):
    progress = tmp_path / "progress.jsonl"
    ckpt_dir = tmp_path / "ckpt"
    env = dict(os.environ)
    env.update(
        {
            "ELASTIC_JOB_NAME": f"e2e{uuid.uuid4().hex[:6]}",
            "DLROVER_IPC_SOCKET_DIR": str(tmp_path / "ipc"),
            "MASTER_ADDR": "127.0.0.1",
            "DLROVER_LOG_LEVEL": "INFO",
        }
    )
    env.update(extra_env or {})
    cmd = [
        sys.executable,
        "-m",
        "dlrover_amd.trainer.elastic_run",
        "--standalone",
        "--nproc-per-node",
        str(nproc),
        "--max-restarts",
        str(max_restarts),
        "--monitor-interval",
        "1",
        "--checkpoint-dir",
        str(ckpt_dir),
        os.path.join(ROOT, "examples", "train_nanogpt.py"),
        "--steps",
        str(steps),
        "--ckpt-interval",
        str(ckpt_interval),
        "--ckpt-dir",
        str(ckpt_dir),
        "--progress-file",
        str(progress),
    ]
    proc = subprocess.run(
        cmd, cwd=ROOT, env=env, capture_output=True, text=True, timeout=timeout
    )
    return proc, progress, ckpt_dir


def _read_progress(progress):
    if not progress.exists():
        return []
    return [json.loads(l) for l in progress.read_text().splitlines() if l.strip()]


@pytest.mark.timeout(600)
def test_nanogpt_ddp_ws2_clean_run(tmp_path):
    proc, progress, ckpt_dir = _run_cli(tmp_path, steps=8, ckpt_interval=4)
    assert proc.returncode == 0, f"stdout:\n{proc.stdout[-3000:]}\nstderr:\n{proc.stderr[-5000:]}"
    rows = _read_progress(progress)
    assert rows and rows[-1]["step"] == 8
    # committed checkpoint on disk
    from dlrover_amd.common.storage import read_tracker_step

    assert read_tracker_step(str(ckpt_dir)) == 8
    assert (ckpt_dir / "8" / "rank_00000.pt").exists()


@pytest.mark.timeout(600)
def test_nanogpt_ddp_ws2_sigkill_recovery(tmp_path):
    proc, progress, ckpt_dir = _run_cli(
        tmp_path,
        extra_env={"DLROVER_TEST_KILL_AT_STEP": "7"},
        steps=12,
        ckpt_interval=3,
    )
    assert proc.returncode == 0, f"stdout:\n{proc.stdout[-3000:]}\nstderr:\n{proc.stderr[-5000:]}"
    rows = _read_progress(progress)
    assert rows and rows[-1]["step"] == 12
    # there must be a second incarnation that resumed from the step-6 ckpt
    incarnations = {r["incarnation"] for r in rows}
    assert 1 in incarnations, f"no restart observed: {rows}"
    resumed = [r for r in rows if r["incarnation"] == 1]
    assert resumed[0]["resumed_from"] == 6, resumed[:2]
    from dlrover_amd.common.storage import read_tracker_step

    assert read_tracker_step(str(ckpt_dir)) == 12
