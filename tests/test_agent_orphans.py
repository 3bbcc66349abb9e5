"""Unit coverage for the agent's exact-PID descendant tracking + orphan
sweep (ref training.py:800 _stop_orphan_workers — HIP
contexts make orphaned dataloader processes pin GPU memory)."""

import os
import subprocess
import sys
import time


def _state(pid: int) -> str:
    """'' if gone, else the /proc state letter ('Z' for zombie)."""
    try:
        with open(f"/proc/{pid}/stat") as f:
            return f.read().split(")")[-1].split()[0]
    except OSError:
        return ""


def test_worker_descendants_and_orphan_sweep():
    from dlrover_amd.agent.training import ElasticTrainingAgent

    # a "worker" that spawns a grandchild (the dataloader analog) and sleeps
    parent = subprocess.Popen(
        [sys.executable, "-c", (
            "import subprocess, sys, time\n"
            "c = subprocess.Popen([sys.executable, '-c', "
            "'import time; time.sleep(300)'])\n"
            "print(c.pid, flush=True)\n"
            "time.sleep(300)\n"
        )],
        stdout=subprocess.PIPE, text=True,
    )
    try:
        grandchild = int(parent.stdout.readline())

        agent = object.__new__(ElasticTrainingAgent)  # no torchelastic init

        class FakePC:
            def pids(self):
                return {0: parent.pid}

        agent._pcontext = FakePC()
        pids = agent._worker_descendants()
        assert parent.pid in pids and grandchild in pids, pids

        # the parent dies (SIGKILL) — the grandchild is orphaned but alive
        parent.kill()
        parent.wait(timeout=10)
        assert _state(grandchild) not in ("", "Z")

        agent._sweep_orphans(pids, grace=1.0)
        deadline = time.time() + 10
        while time.time() < deadline and _state(grandchild) not in ("", "Z"):
            time.sleep(0.1)
        assert _state(grandchild) in ("", "Z"), "orphan survived the sweep"
    finally:
        for pid in (parent.pid,):
            try:
                os.kill(pid, 9)
            except OSError:
                pass


def test_sweep_ignores_already_dead_pids():
    from dlrover_amd.agent.training import ElasticTrainingAgent

    agent = object.__new__(ElasticTrainingAgent)
    # a PID that certainly does not exist; must be a no-op, not an error
    agent._sweep_orphans([2 ** 22 - 3], grace=0.1)
