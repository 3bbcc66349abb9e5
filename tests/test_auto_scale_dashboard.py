"""Auto-scaler plans + dashboard endpoints."""

import json
import urllib.request

import pytest

from dlrover_amd.common.constants import NodeStatus, NodeType
from dlrover_amd.common.node import Node
from dlrover_amd.master.auto_scale import (
    JobAutoScaler,
    LocalResourceOptimizer,
    SimpleStrategyGenerator,
)
from dlrover_amd.master.dashboard import Dashboard
from dlrover_amd.master.job_master import LocalJobMaster
from dlrover_amd.master.node.job_context import JobContext
from dlrover_amd.master.perf_monitor import PerfMonitor
from dlrover_amd.master.scaler.pod_scaler import FakeK8sApi, PodScaler


@pytest.fixture()
def ctx():
    JobContext._reset_for_tests()
    yield JobContext.singleton_instance()
    JobContext._reset_for_tests()


def _add_node(ctx, nid, status):
    n = Node(NodeType.WORKER, nid, status=status)
    ctx.update_node(n)
    return n


def test_optimizer_shrinks_on_pending(ctx):
    perf = PerfMonitor()
    opt = LocalResourceOptimizer(perf, ctx, min_nodes=2, max_nodes=4)
    for i in range(2):
        _add_node(ctx, i, NodeStatus.RUNNING)
    for i in (2, 3):
        _add_node(ctx, i, NodeStatus.PENDING)
    plan = opt.generate_plan()
    assert plan is not None and plan.node_count == 2


def test_optimizer_grows_to_max(ctx):
    perf = PerfMonitor()
    opt = LocalResourceOptimizer(perf, ctx, min_nodes=1, max_nodes=4)
    _add_node(ctx, 0, NodeStatus.RUNNING)
    plan = opt.generate_plan()
    assert plan is not None and plan.node_count == 4


def test_auto_scaler_executes_via_pod_scaler(ctx):
    perf = PerfMonitor()
    opt = LocalResourceOptimizer(perf, ctx, min_nodes=1, max_nodes=3)
    _add_node(ctx, 0, NodeStatus.RUNNING)
    api = FakeK8sApi()
    scaler = PodScaler("j", api=api)
    try:
        auto = JobAutoScaler(opt, scaler)
        plan = auto.execute_once()
        assert plan.node_count == 3
        import time

        t0 = time.time()
        while len(api.created) < 2 and time.time() - t0 < 10:
            time.sleep(0.05)
        assert len(api.created) == 2  # two new pods to reach 3
    finally:
        scaler.stop()


def test_strategy_generator():
    import dlrover_amd.common.comm as comm

    perf = PerfMonitor()
    gen = SimpleStrategyGenerator(perf)
    cfg = gen.generate_parallel_config(0)
    assert cfg.dataloader.num_workers == 0  # no stats: defaults
    perf.report_resource(
        0,
        comm.ResourceStats(
            node_id=0, gpu_stats=[{"index": 0, "used_mb": 1000, "total_mb": 100000}]
        ),
    )
    cfg = gen.generate_parallel_config(0)
    assert cfg.dataloader.num_workers == 4


def test_dashboard_endpoints(ctx):
    master = LocalJobMaster(port=0).prepare()
    dash = Dashboard(master, port=0).start()
    try:
        base = f"http://127.0.0.1:{dash.port}"
        job = json.loads(urllib.request.urlopen(f"{base}/api/job").read())
        assert "stage" in job and "rdzv_round" in job
        nodes = json.loads(urllib.request.urlopen(f"{base}/api/nodes").read())
        assert nodes == []
        page = urllib.request.urlopen(f"{base}/").read().decode()
        assert "dlrover_amd job master" in page
    finally:
        dash.stop()
        master.stop()


def test_strategy_generator_versioned_growth():
    """Concrete versioned batch/lr suggestions from free GPU memory bounded
    by the activation-memory estimate (ref simple_strategy_generator)."""
    from dlrover_amd.common import comm
    from dlrover_amd.master.auto_scale import SimpleStrategyGenerator

    class FakePerf:
        def __init__(self, free_mb):
            self._free = free_mb

        def node_resource(self, node_id):
            class S:
                pass

            s = S()
            s.gpu_stats = [{"total_mb": 294912,
                            "used_mb": 294912 - self._free}]
            return s

    cur = comm.ParallelConfig(
        dataloader=comm.DataLoaderConfig(batch_size=2, version=3),
        optimizer=comm.OptimizerConfig(learning_rate=1e-4, version=3),
    )
    gen = SimpleStrategyGenerator(FakePerf(free_mb=200000))
    out = gen.generate_parallel_config(0, current=cur,
                                       model={"seq_len": 4096,
                                              "n_layers": 32,
                                              "n_heads": 32,
                                              "hidden_size": 4096})
    assert out.dataloader.batch_size > 2
    assert out.dataloader.batch_size <= 8  # bounded growth
    assert out.dataloader.version == 4
    assert out.optimizer.learning_rate > 1e-4  # sqrt scaling applied

    # no headroom -> unchanged config, same version
    gen2 = SimpleStrategyGenerator(FakePerf(free_mb=1000))
    out2 = gen2.generate_parallel_config(0, current=cur)
    assert out2.dataloader.batch_size == 2
    assert out2.dataloader.version == 3


def test_dashboard_events_tail(tmp_path, monkeypatch):
    import json
    import urllib.request

    monkeypatch.setenv("DLROVER_EVENT_DIR", str(tmp_path))
    (tmp_path / "events_agent_1.jsonl").write_text(
        '{"ts": 1.0, "name": "rendezvous", "phase": "begin"}\n'
        '{"ts": 2.0, "name": "rendezvous", "phase": "end"}\n'
    )
    from dlrover_amd.master.dashboard import Dashboard

    class M:
        pass

    dash = Dashboard(M(), port=0, host="127.0.0.1").start()
    try:
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{dash.port}/api/events", timeout=10
        ).read()
        evs = json.loads(body)
        assert len(evs) == 2 and evs[-1]["phase"] == "end"
    finally:
        dash.stop()


def test_optimizer_consults_brain_when_configured(ctx, monkeypatch):
    """optimizeMode=cluster: with DLROVER_BRAIN_ADDR set, the optimizer asks
    the LIVE Brain service for the plan (and reports throughput) before any
    local heuristic; the plan is clamped to [min_nodes, max_nodes]."""
    from dlrover_amd.common.constants import NodeStatus, NodeType
    from dlrover_amd.common.node import Node
    from dlrover_amd.master.auto_scale import LocalResourceOptimizer
    from dlrover_amd.master.brain_service import BrainService
    from dlrover_amd.master.perf_monitor import PerfMonitor
    import time

    svc = BrainService(port=0, host="127.0.0.1").start()
    try:
        monkeypatch.setenv("DLROVER_BRAIN_ADDR", f"127.0.0.1:{svc.port}")
        for i in range(2):
            n = Node(NodeType.WORKER, i)
            n.update_status(NodeStatus.RUNNING)
            ctx.update_node(n)
        perf = PerfMonitor()
        perf.report_global_step(10, time.time() - 1)
        perf.report_global_step(20, time.time())
        opt = LocalResourceOptimizer(perf, ctx=ctx, min_nodes=1, max_nodes=8,
                                     job_name="jbrain")
        plan = opt.generate_plan()
        assert plan is not None and plan.comment == "brain:grow"
        assert plan.node_count == 4  # 2 alive -> brain doubles
    finally:
        svc.stop()


def test_quota_bounds_growth(ctx):
    """Scale-up is clamped by the platform quota (ref: cluster/quota.py):
    zero free capacity means no grow plan even below max_nodes."""
    from dlrover_amd.common.constants import NodeStatus, NodeType
    from dlrover_amd.common.node import Node
    from dlrover_amd.master.auto_scale import (
        LocalResourceOptimizer,
        NoFreeQuotaChecker,
    )
    from dlrover_amd.master.perf_monitor import PerfMonitor

    for i in range(2):
        n = Node(NodeType.WORKER, i)
        n.update_status(NodeStatus.RUNNING)
        ctx.update_node(n)
    opt = LocalResourceOptimizer(PerfMonitor(), ctx=ctx, min_nodes=1,
                                 max_nodes=8, quota=NoFreeQuotaChecker())
    assert opt.generate_plan() is None  # no free nodes -> hold

    class TwoFree:
        def get_free_node_num(self):
            return 2

    opt2 = LocalResourceOptimizer(PerfMonitor(), ctx=ctx, min_nodes=1,
                                  max_nodes=8, quota=TwoFree())
    plan = opt2.generate_plan()
    assert plan is not None and plan.node_count == 4  # 2 alive + 2 free
