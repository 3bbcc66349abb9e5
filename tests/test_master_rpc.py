"""Master <-> client integration over real TCP RPC (single process)."""

import pytest

from dlrover_amd.common import comm
from dlrover_amd.common.constants import NodeEventType, RendezvousName
from dlrover_amd.diagnosis.actions import DiagnosisActionType, NodeAction
from dlrover_amd.agent.master_client import MasterClient
from dlrover_amd.master.job_master import LocalJobMaster
from dlrover_amd.master.node.job_context import JobContext


@pytest.fixture()
def master():
    JobContext._reset_for_tests()
    m = LocalJobMaster(port=0).prepare()
    yield m
    m.stop()
    JobContext._reset_for_tests()


@pytest.fixture()
def client(master):
    c = MasterClient(f"127.0.0.1:{master.port}", node_id=0)
    yield c
    c.close()


def test_rendezvous_two_nodes(master, client):
    client.report_rdzv_params(2, 2, waiting_timeout=30, node_unit=1)
    c1 = MasterClient(f"127.0.0.1:{master.port}", node_id=1)
    client.join_rendezvous(0, 8)
    rnd, group, world = client.get_comm_world(RendezvousName.TRAINING, 0)
    assert world == {}  # only one of two joined
    c1.join_rendezvous(1, 8)
    rnd, group, world = client.get_comm_world(RendezvousName.TRAINING, 0)
    assert world == {0: 8, 1: 8} and rnd == 1
    assert client.num_nodes_waiting() == 0
    c1.close()


def test_rendezvous_node_unit_truncation(master, client):
    client.report_rdzv_params(2, 4, waiting_timeout=0.2, node_unit=2)
    clients = [MasterClient(f"127.0.0.1:{master.port}", node_id=i) for i in range(3)]
    for i, c in enumerate(clients):
        c.join_rendezvous(i, 8)
    import time

    time.sleep(0.4)
    # only 3 of (max) alive... all alive joined => completes with all 3?
    # 3 alive, all joined -> completed, truncated to node_unit=2 -> world {0,1}
    rnd, _, world = clients[0].get_comm_world(RendezvousName.TRAINING, 0)
    assert set(world) == {0, 1}
    # node 2 still waiting for the next round
    assert clients[0].num_nodes_waiting() == 1
    for c in clients:
        c.close()


def test_kv_store(client):
    client.kv_store_set("alpha", b"1")
    assert client.kv_store_get("alpha") == b"1"
    assert client.kv_store_get("missing") == b""
    assert client.kv_store_add("ctr", 5) == 5
    assert client.kv_store_add("ctr", 2) == 7
    client.kv_store_multi_set({"a": b"x", "b": b"y"})
    assert client.kv_store_multi_get(["a", "b"]) == {"a": b"x", "b": b"y"}
    client.kv_store_delete("alpha")
    assert client.kv_store_get("alpha") == b""


def test_heartbeat_carries_action(master, client):
    client.join_rendezvous(0, 8)
    master.ctx.enqueue_action(
        NodeAction(action_type=DiagnosisActionType.RESTART_WORKER, node_id=0, reason="x")
    )
    resp = client.report_heart_beat(0)
    assert resp.action_kwargs["action_type"] == DiagnosisActionType.RESTART_WORKER
    # queue drained
    resp = client.report_heart_beat(0)
    assert resp.action_cls == ""


def test_data_sharding_flow(client):
    client.report_dataset_params(
        comm.DatasetShardParams(dataset_name="ds", dataset_size=10, shard_size=4,
                                num_epochs=1)
    )
    seen = []
    while True:
        task = client.get_task("ds")
        if task.empty:
            break
        seen.append((task.start, task.end))
        client.report_task_result("ds", task.task_id, success=True)
    assert seen == [(0, 4), (4, 8), (8, 10)]


def test_shard_recovery_on_failure(master, client):
    client.report_dataset_params(
        comm.DatasetShardParams(dataset_name="ds2", dataset_size=8, shard_size=4)
    )
    t1 = client.get_task("ds2")
    assert not t1.empty
    # node dies: the failure callback re-queues its in-flight shard
    from dlrover_amd.common import comm as _c
    from dlrover_amd.common.constants import NodeEventType as _NE
    from dlrover_amd.common.constants import NodeType as _NT

    master.job_manager.on_node_event(_c.NodeEvent(
        event_type=_NE.FAILED_EXITED,
        node=_c.NodeMeta(type=_NT.WORKER, id=0, rank=0),
        reason="crash",
    ))
    t2 = client.get_task("ds2")
    assert (t2.start, t2.end) == (t1.start, t1.end)


def test_node_event_and_running_nodes(master, client):
    client.join_rendezvous(0, 8)
    nodes = client.get_running_nodes()
    assert len(nodes) == 1 and nodes[0].id == 0
    client.report_node_event(NodeEventType.SUCCEEDED_EXITED)
    assert master.ctx.is_stopping()


def test_network_check_flow(master):
    clients = [MasterClient(f"127.0.0.1:{master.port}", node_id=i) for i in range(4)]
    for i, c in enumerate(clients):
        c.report_rdzv_params(4, 4, 30, 1)
        c.join_rendezvous(i, 8, rdzv_name=RendezvousName.NETWORK_CHECK)
    # pair groups: (0,1), (2,3)
    _, g0, w0 = clients[0].get_comm_world(RendezvousName.NETWORK_CHECK, 0)
    _, g2, w2 = clients[2].get_comm_world(RendezvousName.NETWORK_CHECK, 2)
    assert set(w0) == {0, 1} and set(w2) == {2, 3}
    # node 1 is slow/fault
    times = {0: 5.0, 1: 120.0, 2: 5.5, 3: 5.2}
    for i, c in enumerate(clients):
        c.report_network_check_result(i, normal=(i != 1), elapsed=times[i])
    stragglers = clients[0].check_straggler()
    assert stragglers == [1]
    fault, reason = clients[0].check_fault_node()
    assert fault == [1]
    for c in clients:
        c.close()


def test_ckpt_sync(master, client):
    client.report_rdzv_params(1, 1, 30, 1)
    client.join_rendezvous(0, 8)
    client.get_comm_world(RendezvousName.TRAINING, 0)
    assert client.sync_checkpoint(50) is True


def test_barrier_and_sync(client):
    client.join_sync("warmup")
    assert not client.is_sync_finished("warmup")
    client.sync_finished("warmup")
    assert client.is_sync_finished("warmup")
    assert not client.barrier("b1")
    assert client.barrier("b1", notify=True)


def test_rdzv_block_over_rpc(master, client):
    """block_rendezvous travels the wire: a held round does not complete,
    release lets it form (ref UcpRdzvManager)."""
    client.report_rdzv_params(1, 4, 0.01, 1)
    client.block_rendezvous(0, True)
    client.join_rendezvous(0, 8)
    import time as _t

    _t.sleep(0.05)
    _, _, world = client.get_comm_world(RendezvousName.TRAINING, 0)
    assert world == {}
    client.block_rendezvous(0, False)
    _, _, world = client.get_comm_world(RendezvousName.TRAINING, 0)
    assert set(world) == {0}
