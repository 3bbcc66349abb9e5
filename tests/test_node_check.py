"""Node health check e2e on CPU: real probe workers (gloo matmul+allreduce)
through the NETWORK_CHECK rendezvous, healthy and fault-injected paths."""

import os
import subprocess
import sys
import uuid

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

SCRIPT = """
import os, sys
from dlrover_amd.agent.master_client import MasterClient
from dlrover_amd.agent.node_check_agent import NodeCheckFailedError, run_network_check
from dlrover_amd.agent.training import ElasticLaunchConfig

os.environ["DLROVER_MASTER_ADDR"] = sys.argv[1]
os.environ["NODE_ID"] = "0"
client = MasterClient.singleton_instance()
client.report_rdzv_params(1, 1, 30, 1)
config = ElasticLaunchConfig(min_nodes=1, max_nodes=1, nproc_per_node=2,
                             node_rank=0, rdzv_timeout=120)
try:
    ok = run_network_check(config, client)
    print("CHECK_RESULT ok" if ok else "CHECK_RESULT bad")
except NodeCheckFailedError as e:
    print(f"CHECK_RESULT failed: {e}")
"""


def _run_check(master_addr, extra_env=None, timeout=300):
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env.update(extra_env or {})
    return subprocess.run(
        [sys.executable, "-c", SCRIPT, master_addr],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=timeout,
    )


@pytest.mark.timeout(600)
def test_node_check_healthy(tmp_path):
    from dlrover_amd.testing import MasterProcess

    env = {"ELASTIC_JOB_NAME": f"nc{uuid.uuid4().hex[:6]}"}
    with MasterProcess(env) as master:
        out = _run_check(master.addr, extra_env=env)
        assert "CHECK_RESULT ok" in out.stdout, (
            out.stdout[-2000:], out.stderr[-3000:]
        )


@pytest.mark.timeout(600)
def test_node_check_mock_fault(tmp_path):
    """MOCK_ERR_RANK makes a probe rank throw (ref: node_check/utils.py:52);
    the single node is then judged a fault node and the agent raises."""
    from dlrover_amd.testing import MasterProcess

    env = {"ELASTIC_JOB_NAME": f"nc{uuid.uuid4().hex[:6]}", "MOCK_ERR_RANK": "0"}
    with MasterProcess(env) as master:
        out = _run_check(master.addr, extra_env=env)
        assert "CHECK_RESULT failed" in out.stdout, (
            out.stdout[-2000:], out.stderr[-3000:]
        )


@pytest.mark.timeout(300)
def test_node_check_straggler_localization():
    """Two nodes, one slow: the master's 2-round pairing flags the slow
    node as a straggler over the real RPC path (the BASELINE-cited
    chaos-experiment behavior, fault_tolerance_exps.md:149-156)."""
    from dlrover_amd.agent.master_client import MasterClient
    from dlrover_amd.testing import MasterProcess

    env = {"ELASTIC_JOB_NAME": f"nc{uuid.uuid4().hex[:6]}"}
    with MasterProcess(env) as master:
        os.environ["DLROVER_MASTER_ADDR"] = master.addr
        clients = [MasterClient(master.addr, node_id=0),
                   MasterClient(master.addr, node_id=1)]
        clients[0].report_rdzv_params(2, 2, 60, 1)
        for round_times in ({0: 20.3, 1: 206.9}, {0: 20.1, 1: 201.8}):
            for n, c in enumerate(clients):
                c.join_rendezvous(n, 1, rdzv_name="network-check")
            for n, c in enumerate(clients):
                # poll until the check round's world forms
                import time as _t

                deadline = _t.time() + 60
                while _t.time() < deadline:
                    _, _, world = c.get_comm_world("network-check", n)
                    if world:
                        break
                    _t.sleep(0.2)
                assert world, f"node {n} never grouped"
            for n, c in enumerate(clients):
                c.report_network_check_result(n, True, round_times[n])
        stragglers = clients[0].check_straggler()
        assert stragglers == [1], stragglers
        faults, _ = clients[0].check_fault_node()
        assert faults == []  # slow, not broken
