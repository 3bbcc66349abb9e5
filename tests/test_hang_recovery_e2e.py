"""Config #5 plumbing on CPU: one rank stalls mid-collective -> the master's
hang diagnostician (no global-step progress past hang_downtime) pushes a
RESTART_WORKER action on heartbeats -> the agent restarts workers -> training
resumes from the flash checkpoint and completes.

On GPU the same ladder triggers from hiptimer's XPU_TIMER_COMMON_HANG metric
(a stalled RCCL all-reduce keeps outstanding events from completing); the
hiptimer hang signal is unit-tested in test_hiptimer.py.
"""

import json
import os
import subprocess
import sys
import uuid

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_hang_detect_and_restart(tmp_path):
    progress = tmp_path / "progress.jsonl"
    ckpt_dir = tmp_path / "ckpt"
    env = dict(os.environ)
    env.update(
        {
            "ELASTIC_JOB_NAME": f"hang{uuid.uuid4().hex[:6]}",
            "DLROVER_IPC_SOCKET_DIR": str(tmp_path / "ipc"),
            "MASTER_ADDR": "127.0.0.1",
            "DLROVER_TEST_HANG_AT_STEP": "6",
            "DLROVER_HANG_DOWNTIME": "8",
            "DLROVER_PG_TIMEOUT": "600",  # the gloo timeout must NOT fire:
            # recovery must come from the hang diagnostician, not the pg
            "DLROVER_LOG_LEVEL": "INFO",
        }
    )
    cmd = [
        sys.executable, "-m", "dlrover_amd.trainer.elastic_run",
        "--standalone", "--nproc-per-node", "2",
        "--max-restarts", "2", "--monitor-interval", "1",
        "--checkpoint-dir", str(ckpt_dir),
        os.path.join(ROOT, "examples", "train_nanogpt.py"),
        "--steps", "12", "--ckpt-interval", "3",
        "--ckpt-dir", str(ckpt_dir),
        "--progress-file", str(progress),
    ]
    proc = subprocess.run(cmd, cwd=ROOT, env=env, capture_output=True,
                          text=True, timeout=420)
    assert proc.returncode == 0, proc.stderr[-5000:]
    rows = [json.loads(l) for l in progress.read_text().splitlines() if l.strip()]
    assert rows[-1]["step"] == 12
    # a second incarnation resumed from the last pre-hang checkpoint
    incarnations = {r["incarnation"] for r in rows}
    assert 1 in incarnations, rows[-5:]
    resumed = [r for r in rows if r["incarnation"] == 1]
    assert resumed[0]["resumed_from"] == 3, resumed[:2]
    assert "hang" in proc.stderr or "hang" in proc.stdout
