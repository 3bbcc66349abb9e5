"""PodScaler + PodWatcher against the fake k8s API (the reference's test
pattern: tests/test_utils.py mocked k8sClient)."""

import time

import pytest

from dlrover_amd.common.constants import NodeEventType, NodeStatus, NodeType
from dlrover_amd.common.events import AsyncExporter, EventEmitter
from dlrover_amd.common.node import Node, NodeResource
from dlrover_amd.master.scaler.pod_scaler import (
    FakeK8sApi,
    PodScaler,
    build_pod_spec,
)
from dlrover_amd.master.watcher.k8s_watcher import (
    FakeEventSource,
    PodWatcher,
    pod_to_node_event,
)


def _wait(cond, timeout=10):
    t0 = time.time()
    while time.time() - t0 < timeout:
        if cond():
            return True
        time.sleep(0.05)
    return False


def test_pod_spec_has_amd_gpu_and_env():
    node = Node(NodeType.WORKER, 3, config_resource=NodeResource(gpu_num=8))
    spec = build_pod_spec("jobx", node, master_addr="10.0.0.1:24666")
    limits = spec["spec"]["containers"][0]["resources"]["limits"]
    assert limits["amd.com/gpu"] == "8"
    env = {e["name"]: e["value"] for e in spec["spec"]["containers"][0]["env"]}
    assert env["DLROVER_MASTER_ADDR"] == "10.0.0.1:24666"
    assert env["HSA_ENABLE_IPC_MODE_LEGACY"] == "0"
    assert spec["metadata"]["labels"]["elasticjob.dlrover/replica-index"] == "3"


def test_scaler_launch_and_remove():
    api = FakeK8sApi()
    scaler = PodScaler("jobx", api=api)
    try:
        node = Node(NodeType.WORKER, 0, config_resource=NodeResource(gpu_num=8))
        scaler.launch_node(node)
        assert _wait(lambda: "jobx-worker-0" in api.pods)
        scaler.remove_node(node)
        assert "jobx-worker-0" in api.deleted
    finally:
        scaler.stop()


def test_scaler_scale_to():
    api = FakeK8sApi()
    scaler = PodScaler("jobx", api=api)
    try:
        nodes = [Node(NodeType.WORKER, i, status=NodeStatus.RUNNING) for i in range(2)]
        scaler.scale_to(4, nodes)
        assert _wait(lambda: len(api.created) == 2)
        # scale down removes highest ranks
        all_nodes = nodes + [
            Node(NodeType.WORKER, i, status=NodeStatus.RUNNING) for i in (2, 3)
        ]
        scaler.scale_to(2, all_nodes)
        assert _wait(lambda: len(api.deleted) == 2)
        assert set(api.deleted) == {"jobx-worker-3", "jobx-worker-2"}
    finally:
        scaler.stop()


def test_scaler_scale_up_refills_rank_holes():
    """After a scale-down removed ranks 2-3, scaling back up must re-fill
    ranks 2-3 (contiguous 0..count-1), not extend to 4-5."""
    api = FakeK8sApi()
    scaler = PodScaler("jobx", api=api)
    try:
        nodes = [
            Node(NodeType.WORKER, i, status=NodeStatus.RUNNING) for i in range(4)
        ]
        for dead in nodes[2:]:
            dead.status = NodeStatus.FAILED
        scaler.scale_to(4, nodes)
        assert _wait(lambda: len(api.created) == 2)
        ranks = sorted(
            int(api.pods[name]["metadata"]["labels"][
                "elasticjob.dlrover/rank-index"])
            for name in api.created
        )
        assert ranks == [2, 3]
        # ids still grow monotonically (no pod-name reuse)
        assert set(api.created) == {"jobx-worker-4", "jobx-worker-5"}
    finally:
        scaler.stop()


def _pod(idx, phase, reason=""):
    pod = {
        "metadata": {
            "name": f"j-worker-{idx}",
            "labels": {
                "elasticjob.dlrover/name": "j",
                "elasticjob.dlrover/replica-type": "worker",
                "elasticjob.dlrover/replica-index": str(idx),
                "elasticjob.dlrover/rank-index": str(idx),
            },
        },
        "status": {"phase": phase, "podIP": "10.1.1.5"},
    }
    if reason:
        pod["status"]["containerStatuses"] = [
            {"state": {"terminated": {"reason": reason, "exitCode": 137}}}
        ]
    return pod


def test_pod_event_translation():
    evt = pod_to_node_event("MODIFIED", _pod(2, "Failed", reason="OOMKilled"))
    assert evt.node.id == 2 and evt.node.status == NodeStatus.FAILED
    assert evt.reason == "OOMKilled"
    evt = pod_to_node_event("DELETED", _pod(1, "Running"))
    assert evt.event_type == NodeEventType.DELETED


def test_watcher_stream():
    src = FakeEventSource()
    watcher = PodWatcher("j", source=src)
    src.push("ADDED", _pod(0, "Pending"))
    src.push("MODIFIED", _pod(0, "Running"))
    events = list(watcher.watch())
    assert [e.node.status for e in events] == [NodeStatus.PENDING, NodeStatus.RUNNING]


def test_events_exporter(tmp_path, monkeypatch):
    monkeypatch.setenv("DLROVER_EVENT_DIR", str(tmp_path))
    exporter = AsyncExporter(path=str(tmp_path / "ev.jsonl"))
    em = EventEmitter("test", exporter)
    em.instant("node_join", {"rank": 1})
    with em.duration("rendezvous", {"round": 2}):
        time.sleep(0.01)
    exporter.close()
    import json

    rows = [json.loads(l) for l in open(tmp_path / "ev.jsonl")]
    assert [r["type"] for r in rows] == ["instant", "begin", "end"]
    assert rows[2]["duration_s"] > 0


def test_scaleplan_watcher_converts_manual_plans():
    """User-submitted ScalePlan CR -> ResourcePlan (ref k8s_watcher.py:354);
    only ADDED manual plans for this job, each uid once."""
    from dlrover_amd.master.watcher.k8s_watcher import (
        FakeEventSource,
        ScalePlanWatcher,
    )

    src = FakeEventSource()
    crd = {
        "kind": "ScalePlan",
        "metadata": {
            "name": "sp1",
            "uid": "u1",
            "labels": {"elasticjob.dlrover/name": "jobA",
                       "scale-type": "manual"},
        },
        "spec": {"replicaResourceSpecs": {
            "worker": {"replicas": 6, "resource": {"cpu": "8", "memory": "1024"}}
        }},
    }
    src.push("ADDED", crd)
    src.push("ADDED", crd)  # duplicate uid -> ignored
    src.push("MODIFIED", dict(crd, metadata={"name": "sp1", "uid": "u2",
                                             "labels": crd["metadata"]["labels"]}))
    other = {
        "kind": "ScalePlan",
        "metadata": {"name": "spB", "uid": "u3",
                     "labels": {"elasticjob.dlrover/name": "jobB"}},
        "spec": {"replicaResourceSpecs": {"worker": {"replicas": 2}}},
    }
    src.push("ADDED", other)  # different job -> ignored
    w = ScalePlanWatcher("jobA", source=src)
    plans = list(w.watch())
    assert len(plans) == 1
    assert plans[0].node_count == 6
    assert plans[0].node_resource["cpu"] == 8.0
    assert plans[0].comment == "scaleplan/sp1"


def test_crd_manifests_parse():
    import os

    import yaml

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for f in ("elasticjob-crd.yaml", "scaleplan-crd.yaml",
              "example-elasticjob.yaml"):
        docs = list(yaml.safe_load_all(open(os.path.join(root, "deploy/k8s", f))))
        assert docs and docs[0].get("kind") in (
            "CustomResourceDefinition", "ElasticJob",
        ), f


def test_service_per_pod():
    """service-per-pod gives workers relaunch-stable DNS names (ref
    pod_scaler.py:776)."""
    import time

    from dlrover_amd.common.node import Node, NodeResource
    from dlrover_amd.common.constants import NodeType
    from dlrover_amd.master.scaler.pod_scaler import FakeK8sApi, PodScaler

    api = FakeK8sApi()
    sc = PodScaler("jobx", api=api, service_per_pod=True)
    try:
        n = Node(NodeType.WORKER, 3, rank_index=3,
                 config_resource=NodeResource(gpu_num=8))
        sc.launch_node(n)
        deadline = time.time() + 10
        while time.time() < deadline and not api.services:
            time.sleep(0.05)
        assert "jobx-worker-3" in api.services
        svc = api.services["jobx-worker-3"]
        assert svc["spec"]["selector"]["elasticjob.dlrover/rank-index"] == "3"
        assert svc["spec"]["clusterIP"] == "None"
    finally:
        sc.stop()


def test_elasticjob_watcher_replicas_and_suspend():
    """ElasticJob CR edits drive the job (ref: K8sElasticJobWatcher):
    replica change -> scale plan; suspend toggles -> suspend/resume
    signals; the initial spec is not a change."""
    from dlrover_amd.master.watcher.k8s_watcher import (
        ElasticJobWatcher,
        FakeEventSource,
    )

    def cr(replicas, suspend=False):
        return {
            "kind": "ElasticJob",
            "metadata": {"name": "jobx"},
            "spec": {
                "suspend": suspend,
                "replicaSpecs": {"worker": {"replicas": replicas}},
            },
        }

    src = FakeEventSource()
    w = ElasticJobWatcher("jobx", source=src)
    src.push("ADDED", cr(4))            # initial spec: no event
    src.push("MODIFIED", cr(8))         # scale up
    src.push("MODIFIED", cr(8, True))   # suspend
    src.push("MODIFIED", cr(2, False))  # resume + scale down
    src.push("MODIFIED", {"kind": "ElasticJob",
                          "metadata": {"name": "other"},
                          "spec": {"replicaSpecs": {"worker": {"replicas": 1}}}})
    events = list(w.watch())
    kinds = [(k, p.node_count if p else None) for k, p in events]
    assert kinds == [("scale", 8), ("suspend", None), ("resume", None),
                     ("scale", 2)], kinds
