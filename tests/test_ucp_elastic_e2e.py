"""UCP end-to-end through the AGENT path: FSDP tiny
Llama under two elastic agents; world size changes 1 -> 2 -> 1 mid-job and
every transition resumes by UCP-resharding the sharded checkpoint (model +
optimizer state), driven by the agent's membership-change restart (which
persists all shm shards to disk first) and the worker-side reshard-on-load
(flash_checkpoint/fsdp.py load_checkpoint -> ucp.load_resharded).

Ref behavior: training.py:1548-1651 UCP hook + UcpRdzvManager
(rdzv_manager.py:583) — the reference blocks rendezvous and converts; ours
persists at the agent then reshards in parallel at load."""

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
        os.path.join(ROOT, "examples", "train_llama_fsdp.py"),
        "--model", "tiny",
        "--seq", "64",
        "--steps", str(steps),
        "--ckpt-interval", "3",
        "--ckpt-dir", str(tmp_path / "ckpt"),
        "--progress-file", str(progress),
    ]
    env = dict(env)
    env["NODE_ID"] = str(node_rank)
    env["DLROVER_IPC_SOCKET_DIR"] = str(tmp_path / f"ipc{node_rank}")
    log = open(tmp_path / f"agent{node_rank}.log", "w")
    return subprocess.Popen(cmd, cwd=ROOT, env=env, stdout=log,
                            stderr=subprocess.STDOUT, text=True)


def _rows(progress):
    if not os.path.exists(progress):
        return []
    return [json.loads(l) for l in open(progress).read().splitlines() if l.strip()]


@pytest.mark.timeout(900)
def test_fsdp_ucp_scale_up_down_e2e(tmp_path):
    env = dict(os.environ)
    env.update(
        {
            "ELASTIC_JOB_NAME": f"ucp{uuid.uuid4().hex[:6]}",
            "MASTER_ADDR": "127.0.0.1",
            "DLROVER_FSDP_CPU": "1",
            "DLROVER_PG_TIMEOUT": "25",
            "DLROVER_HEARTBEAT_TIMEOUT": "15",
            "DLROVER_LOG_LEVEL": "INFO",
        }
    )
    progress = tmp_path / "progress.jsonl"
    steps = 600
    master = a0 = a1 = None
    try:
        master, addr = _spawn_master(env)
        a0 = _spawn_agent(env, addr, 0, tmp_path, steps, progress)
        # phase 1: ws=1 trains past a disk checkpoint (interval 3)
        deadline = time.time() + 180
        while time.time() < deadline:
            if any(r["step"] >= 5 for r in _rows(progress)):
                break
            time.sleep(1)
        assert _rows(progress), "node 0 never trained"

        # phase 2: node 1 joins -> membership restart -> ws=2 resumes by
        # UCP-resharding the ws-1 shards (grow direction)
        a1 = _spawn_agent(env, addr, 1, tmp_path, steps, progress)
        deadline = time.time() + 240
        up_row = None
        while time.time() < deadline:
            rows = _rows(progress)
            if rows and rows[-1].get("world") == 2:
                up_row = rows[-1]
                break
            if a0.poll() is not None:
                break
            time.sleep(1)
        assert up_row is not None, f"no scale-up: {_rows(progress)[-3:]}"
        # the resharded resume continued from the persisted step, not 0
        ws2 = [r for r in _rows(progress) if r.get("world") == 2]
        assert ws2 and ws2[0].get("resumed_from", 0) >= 3, ws2[:2]

        # phase 3: node 1 dies -> ws=1 again, resharding 2 -> 1
        a1.terminate()
        a1.wait(timeout=30)
        a0.wait(timeout=420)
        assert a0.returncode == 0, open(tmp_path / "agent0.log").read()[-4000:]
        rows = _rows(progress)
        assert rows[-1]["step"] == steps, rows[-5:]
        worlds = {r.get("world") for r in rows}
        assert worlds == {1, 2}, worlds
        # the final ws-1 incarnation resumed from a mid-job step (UCP 2->1
        # with optimizer state — FusedAdamW would raise on shape mismatch)
        down = [r for r in rows if r.get("world") == 1 and r.get("resumed_from", 0) > 0]
        assert down, rows[-5:]
    finally:
        for p in (a0, a1, master):
            if p is not None and p.poll() is None:
                p.terminate()
                try:
                    p.wait(timeout=20)
                except subprocess.TimeoutExpired:
                    p.kill()
