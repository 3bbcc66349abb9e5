"""Config #4 plumbing on CPU: elastic scale 1 -> 2 -> 1 NODES mid-job with
re-rendezvous, through the full production stack (master subprocess, two real
elastic agents, gloo workers, checkpoint-resume across world changes).

The GPU version (4->8->4 over RCCL) runs the same code paths with
nproc-per-node GPUs; here each "node" contributes 1 CPU worker.
"""

import json
import os
import subprocess
import sys
import tempfile
import time
import uuid

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _spawn_master(env):
    port_file = os.path.join(tempfile.gettempdir(), f"m_{uuid.uuid4().hex[:8]}.port")
    proc = subprocess.Popen(
        [sys.executable, "-m", "dlrover_amd.master.main", "--platform", "local",
         "--port", "0", "--port_file", port_file],
        env=env,
    )
    deadline = time.time() + 60
    while time.time() < deadline:
        if os.path.exists(port_file):
            content = open(port_file).read().strip()
            if content:
                return proc, f"127.0.0.1:{content}"
        time.sleep(0.2)
    proc.terminate()
    raise RuntimeError("master did not start")


def _spawn_agent(env, master_addr, node_rank, tmp_path, steps, progress):
    cmd = [
        sys.executable, "-m", "dlrover_amd.trainer.elastic_run",
        "--rdzv-endpoint", master_addr,
        "--nnodes", "1:2",
        "--nproc-per-node", "1",
        "--node-rank", str(node_rank),
        "--max-restarts", "3",
        "--monitor-interval", "1",
        "--waiting-timeout", "5",
        "--checkpoint-dir", str(tmp_path / "ckpt"),
        os.path.join(ROOT, "examples", "train_nanogpt.py"),
        "--steps", str(steps),
        "--ckpt-interval", "2",
        "--ckpt-dir", str(tmp_path / "ckpt"),
        "--progress-file", str(progress),
    ]
    env = dict(env)
    env["NODE_ID"] = str(node_rank)
    env["DLROVER_IPC_SOCKET_DIR"] = str(tmp_path / f"ipc{node_rank}")
    # log to a FILE: an unread stdout PIPE fills and blocks the agent
    log = open(tmp_path / f"agent{node_rank}.log", "w")
    return subprocess.Popen(cmd, cwd=ROOT, env=env, stdout=log,
                            stderr=subprocess.STDOUT, text=True)


def _rows(progress):
    if not os.path.exists(progress):
        return []
    return [json.loads(l) for l in open(progress).read().splitlines() if l.strip()]


@pytest.mark.timeout(900)
def test_scale_up_down_e2e(tmp_path):
    env = dict(os.environ)
    env.update(
        {
            "ELASTIC_JOB_NAME": f"scale{uuid.uuid4().hex[:6]}",
            "MASTER_ADDR": "127.0.0.1",
            "DLROVER_PG_TIMEOUT": "25",
            "DLROVER_HEARTBEAT_TIMEOUT": "15",
            "DLROVER_LOG_LEVEL": "INFO",
        }
    )
    progress = tmp_path / "progress.jsonl"
    steps = 1200
    master = None
    a0 = a1 = None
    try:
        master, addr = _spawn_master(env)
        a0 = _spawn_agent(env, addr, 0, tmp_path, steps, progress)
        # phase 1: single node trains
        deadline = time.time() + 120
        while time.time() < deadline:
            if any(r["step"] >= 5 for r in _rows(progress)):
                break
            time.sleep(1)
        assert _rows(progress), "node 0 never trained"

        # phase 2: node 1 joins -> running agent restarts into world of 2
        a1 = _spawn_agent(env, addr, 1, tmp_path, steps, progress)
        deadline = time.time() + 180
        scaled_up = False
        while time.time() < deadline:
            rows = _rows(progress)
            # membership restarts do not consume the restart budget, so the
            # signal is the WORLD growing to 2
            if rows and rows[-1].get("world", 1) == 2:
                scaled_up = True
                break
            if a0.poll() is not None:
                break
            time.sleep(1)
        assert scaled_up, f"no scale-up observed: {_rows(progress)[-3:]}"

        # phase 3: node 1 dies -> collectives fail -> node 0 recovers alone
        a1.terminate()
        a1.wait(timeout=30)
        a0.wait(timeout=420)
        assert a0.returncode == 0, open(tmp_path / "agent0.log").read()[-4000:]
        rows = _rows(progress)
        assert rows[-1]["step"] == steps, rows[-5:]
        worlds = {r.get("world", 1) for r in rows}
        assert worlds == {1, 2}, worlds  # scaled up AND back down
    finally:
        for p in (a0, a1, master):
            if p is not None and p.poll() is None:
                p.terminate()
                try:
                    p.wait(timeout=20)
                except subprocess.TimeoutExpired:
                    p.kill()
