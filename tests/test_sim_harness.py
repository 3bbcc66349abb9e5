"""Sim-master + fake-agent harness scenarios (the reference's testing/
pattern): normal completion and heartbeat-death shrink."""

import time
import uuid

import pytest

from dlrover_amd.testing import FakeAgent, MasterProcess


@pytest.mark.timeout(300)
def test_normal_scenario(tmp_path):
    env = {"ELASTIC_JOB_NAME": f"sim{uuid.uuid4().hex[:6]}"}
    with MasterProcess(env) as master:
        agents = [FakeAgent(master.addr, r, nproc=8) for r in range(2)]
        agents[0].client.report_rdzv_params(2, 2, 30, 1)
        for a in agents:
            a.join()
        worlds = [a.wait_world() for a in agents]
        assert all(set(w) == {0, 1} for w in worlds)
        for a in agents:
            a.report_success()
            a.stop()
        # master reaches SUCCEEDED and exits 0
        deadline = time.time() + 30
        while master.proc.poll() is None and time.time() < deadline:
            time.sleep(0.5)
        assert master.proc.poll() == 0


@pytest.mark.timeout(300)
def test_dead_node_shrinks_world(tmp_path):
    env = {
        "ELASTIC_JOB_NAME": f"sim{uuid.uuid4().hex[:6]}",
        "DLROVER_HEARTBEAT_TIMEOUT": "6",
    }
    with MasterProcess(env) as master:
        a0 = FakeAgent(master.addr, 0).start_heartbeats(interval=1.0)
        a1 = FakeAgent(master.addr, 1).start_heartbeats(interval=1.0)
        a0.client.report_rdzv_params(1, 2, 30, 1)
        a0.join()
        a1.join()
        assert set(a0.wait_world()) == {0, 1}
        # node 1 stops heartbeating (simulated death)
        a1.stop()
        time.sleep(10)  # > heartbeat timeout
        # re-rendezvous: only node 0 in the next world
        a0.join()
        world = a0.wait_world(timeout=60)
        assert set(world) == {0}
        a0.stop()
