"""HTTP transport parity, ElasticJob CRD scaler, MasterKVStore, monitors."""

import time

import pytest

from dlrover_amd.agent.master_client import MasterClient
from dlrover_amd.common.comm import BaseRequest, BaseResponse
from dlrover_amd.common.constants import CommServiceType, NodeStatus, NodeType
from dlrover_amd.common.node import Node
from dlrover_amd.master.job_master import LocalJobMaster
from dlrover_amd.master.node.job_context import JobContext
from dlrover_amd.master.scaler.elasticjob_scaler import ElasticJobScaler, FakeCrdApi
from dlrover_amd.utils.transport import (
    HttpRpcClient,
    HttpRpcServer,
    TcpRpcClient,
    TcpRpcServer,
    wait_for_server,
)


def _echo_handler(verb, req):
    return BaseResponse(success=True, data=req.data)


@pytest.mark.parametrize("server_cls,client_cls,stype", [
    (TcpRpcServer, TcpRpcClient, "tcp"),
    (HttpRpcServer, HttpRpcClient, "http"),
])
def test_transport_roundtrip(server_cls, client_cls, stype):
    from dlrover_amd.common import comm

    server = server_cls(0, _echo_handler)
    server.start()
    try:
        addr = f"127.0.0.1:{server.port}"
        assert wait_for_server(addr, timeout=10, service_type=stype)
        client = client_cls(addr)
        msg = comm.KeyValuePair(key="k", value=b"v" * 100000)  # big payload
        resp = client.call("get", BaseRequest(node_id=1, data=msg))
        assert resp.success and resp.data == msg
        client.close()
    finally:
        server.stop()


def test_master_over_http():
    JobContext._reset_for_tests()
    m = LocalJobMaster(port=0, service_type=CommServiceType.HTTP).prepare()
    try:
        c = MasterClient(f"127.0.0.1:{m.port}", node_id=0,
                         service_type=CommServiceType.HTTP)
        c.kv_store_set("x", b"1")
        assert c.kv_store_get("x") == b"1"
        assert c.join_rendezvous(0, 8) >= 0
        c.close()
    finally:
        m.stop()
        JobContext._reset_for_tests()


def test_master_kv_store_as_torch_store():
    from datetime import timedelta

    from dlrover_amd.agent.master_kv_store import MasterKVStore

    JobContext._reset_for_tests()
    m = LocalJobMaster(port=0).prepare()
    try:
        c = MasterClient(f"127.0.0.1:{m.port}", node_id=0)
        store = MasterKVStore("test", client=c, timeout=timedelta(seconds=5))
        store.set("a", b"hello")
        assert store.get("a") == b"hello"
        assert store.add("ctr", 3) == 3
        assert store.add("ctr", 2) == 5
        store.wait(["a"])
        assert store.check(["a"]) and not store.check(["nope"])
        with pytest.raises(LookupError):
            MasterKVStore("test", client=c, timeout=timedelta(seconds=1)).get("missing")
        c.close()
    finally:
        m.stop()
        JobContext._reset_for_tests()


def test_elasticjob_scaler_emits_scaleplans():
    api = FakeCrdApi()
    scaler = ElasticJobScaler("jobz", api=api)
    node = Node(NodeType.WORKER, 2, status=NodeStatus.RUNNING)
    scaler.launch_node(node)
    scaler.remove_node(node)
    nodes = [Node(NodeType.WORKER, i, status=NodeStatus.RUNNING) for i in range(4)]
    scaler.scale_to(2, nodes)
    assert len(api.plans) == 3
    assert api.plans[0]["spec"]["createdPods"][0]["name"] == "jobz-worker-2"
    assert api.plans[1]["spec"]["removePods"] == ["jobz-worker-2"]
    assert sorted(api.plans[2]["spec"]["removePods"]) == [
        "jobz-worker-2", "jobz-worker-3"
    ]
    assert api.plans[2]["spec"]["replicaResourceSpecs"]["worker"]["replicas"] == 2


def test_resource_monitor_snapshot():
    from dlrover_amd.agent.monitor import ResourceMonitor

    snap = ResourceMonitor().snapshot()
    assert snap.memory_mb > 0


def test_master_over_grpc():
    from dlrover_amd.utils.transport import GrpcRpcClient, GrpcRpcServer

    JobContext._reset_for_tests()
    m = LocalJobMaster(port=0, service_type=CommServiceType.GRPC).prepare()
    try:
        c = MasterClient(f"127.0.0.1:{m.port}", node_id=0,
                         service_type=CommServiceType.GRPC)
        c.kv_store_set("g", b"42")
        assert c.kv_store_get("g") == b"42"
        assert c.join_rendezvous(0, 8) >= 0
        c.close()
    finally:
        m.stop()
        JobContext._reset_for_tests()
