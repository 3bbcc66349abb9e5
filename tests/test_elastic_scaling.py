"""Elastic scale up/down through the master rendezvous (config #4 plumbing):
2 -> 4 -> 2 node worlds with re-rendezvous, as the agents drive it."""

import pytest

from dlrover_amd.agent.master_client import MasterClient
from dlrover_amd.common.constants import RendezvousName
from dlrover_amd.master.job_master import LocalJobMaster
from dlrover_amd.master.node.job_context import JobContext


@pytest.fixture()
def master():
    JobContext._reset_for_tests()
    m = LocalJobMaster(port=0).prepare()
    yield m
    m.stop()
    JobContext._reset_for_tests()


def _join_all(clients, ranks):
    for r in ranks:
        clients[r].join_rendezvous(r, 8)


def test_scale_up_then_down(master):
    clients = {i: MasterClient(f"127.0.0.1:{master.port}", node_id=i) for i in range(4)}
    clients[0].report_rdzv_params(2, 4, waiting_timeout=600, node_unit=2)

    # phase 1: nodes 0,1 form a world of 2
    _join_all(clients, [0, 1])
    _, _, world = clients[0].get_comm_world(RendezvousName.TRAINING, 0)
    assert set(world) == {0, 1}

    # phase 2: nodes 2,3 join -> running agents observe waiting>0
    _join_all(clients, [2, 3])
    assert clients[0].num_nodes_waiting() > 0
    # agents restart workers and re-join; new world = 4 nodes
    _join_all(clients, [0, 1])
    _, _, world = clients[2].get_comm_world(RendezvousName.TRAINING, 2)
    assert set(world) == {0, 1, 2, 3}

    # phase 3: nodes 2,3 die (scale down); master removes them
    for r in (2, 3):
        master.rdzv_managers[RendezvousName.TRAINING].remove_alive_node(r)
    _join_all(clients, [0, 1])
    _, _, world = clients[0].get_comm_world(RendezvousName.TRAINING, 0)
    assert set(world) == {0, 1}
    for c in clients.values():
        c.close()


def test_scale_respects_node_unit(master):
    clients = {i: MasterClient(f"127.0.0.1:{master.port}", node_id=i) for i in range(3)}
    clients[0].report_rdzv_params(2, 4, waiting_timeout=600, node_unit=2)
    _join_all(clients, [0, 1, 2])
    _, _, world = clients[0].get_comm_world(RendezvousName.TRAINING, 0)
    # 3 alive but node_unit=2: world truncated to 2 nodes
    assert set(world) == {0, 1}
    for c in clients.values():
        c.close()


def test_group_network_check_pairing_and_link_suspects():
    """Node-group-aware check: intra-group pairs in even rounds, cross-group
    pairs in odd rounds; a cross-round-only failure implicates the group
    LINK (ref GroupNodeNetworkCheckRendezvousManager)."""
    from dlrover_amd.master.elastic.rdzv_manager import (
        GroupNetworkCheckRendezvousManager,
    )

    mgr = GroupNetworkCheckRendezvousManager()
    # 2 groups x 4 nodes
    mgr._rdzv_nodes = {r: 1 for r in range(8)}
    mgr.set_node_groups({r: r // 4 for r in range(8)})

    pairs0 = mgr._build_groups_locked()
    for p in pairs0:
        gids = {r // 4 for r in p}
        assert len(gids) == 1, f"round-0 pair crosses groups: {p}"

    # everyone passes round 0 quickly
    for r in range(8):
        mgr.report_network_check_result(r, True, 1.0)
    assert mgr._check_round == 1

    pairs1 = mgr._build_groups_locked()
    crossing = [p for p in pairs1 if len({r // 4 for r in p}) == 2]
    assert crossing, pairs1

    # round 1: node 5 fails its cross-group probe
    for r in range(8):
        mgr.report_network_check_result(r, r != 5, 1.0)
    assert (0, 1) in mgr.suspect_group_links()


def test_lastcall_timeout_truncates_to_node_unit(monkeypatch):
    """5 nodes waiting with node_unit=2 at last-call timeout -> a world of 4
    (truncate, ref rdzv_manager.py:183), not a stall."""
    import time as _t

    from dlrover_amd.master.elastic.rdzv_manager import (
        ElasticTrainingRendezvousManager,
    )

    mgr = ElasticTrainingRendezvousManager()
    mgr.update_rdzv_params(2, 8, waiting_timeout=0.2, node_unit=2)
    for r in range(8):
        mgr.add_alive_node(r)  # 8 alive, only 5 will join this round
    for r in range(5):
        mgr.join_rendezvous(r, 1)
    # 5 of 8 alive joined: must wait for the last call
    _, _, world = mgr.get_comm_world(0)
    assert world == {}
    _t.sleep(0.3)
    _, _, world = mgr.get_comm_world(0)
    assert sorted(world) == [0, 1, 2, 3], world  # truncated, rank 4 dropped
    assert mgr.num_nodes_waiting() == 1  # the dropped node waits for next round


def test_blockable_rendezvous_holds_round_open():
    """A UCP-persist hold keeps the pending round from completing — even
    past the last-call timeout — until every holder releases (or dies);
    ref UcpRdzvManager rdzv_manager.py:583."""
    import time as _time

    from dlrover_amd.master.elastic.rdzv_manager import (
        ElasticTrainingRendezvousManager,
    )

    mgr = ElasticTrainingRendezvousManager()
    mgr.update_rdzv_params(1, 4, waiting_timeout=0.01, node_unit=1)
    mgr.block_rendezvous(0, True)
    mgr.join_rendezvous(0, 8)
    mgr.join_rendezvous(1, 8)
    _time.sleep(0.05)  # well past last-call
    rnd, _, world = mgr.get_comm_world(0)
    assert world == {}, "blocked round must not complete"
    # a second holder: releasing only one keeps the hold
    mgr.block_rendezvous(1, True)
    mgr.block_rendezvous(0, False)
    _, _, world = mgr.get_comm_world(0)
    assert world == {}
    mgr.block_rendezvous(1, False)
    _, _, world = mgr.get_comm_world(0)
    assert set(world) == {0, 1}


def test_blockable_rendezvous_dead_holder_released():
    """A holder that dies must not wedge the round forever: the job
    manager's remove_alive_node clears its hold."""
    from dlrover_amd.master.elastic.rdzv_manager import (
        ElasticTrainingRendezvousManager,
    )

    mgr = ElasticTrainingRendezvousManager()
    mgr.update_rdzv_params(1, 4, waiting_timeout=600, node_unit=1)
    mgr.block_rendezvous(3, True)
    mgr.join_rendezvous(0, 8)
    _, _, world = mgr.get_comm_world(0)
    assert world == {}
    mgr.remove_alive_node(3)  # holder died
    _, _, world = mgr.get_comm_world(0)
    assert set(world) == {0}
