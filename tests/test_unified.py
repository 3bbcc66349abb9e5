"""Unified MPMD API: builder validation, local execution, role failover."""

import os
import uuid

import pytest

from dlrover_amd.unified import DLJobBuilder
from dlrover_amd.unified.master import PrimeMaster


def _ok_worker(tag):
    import os

    path = f"/tmp/dlrover_unified_{tag}_{os.environ['ROLE']}_{os.environ['RANK']}"
    with open(path, "w") as f:
        f.write("done")


def _flaky_worker(tag):
    import os

    marker = f"/tmp/dlrover_unified_flaky_{tag}_{os.environ['RANK']}"
    if not os.path.exists(marker):
        with open(marker, "w") as f:
            f.write("x")
        raise SystemExit(3)  # first incarnation dies


def test_builder_validates():
    with pytest.raises(ValueError):
        DLJobBuilder().build()
    with pytest.raises(ValueError):
        DLJobBuilder().role("t").total(2).build()  # no entrypoint


@pytest.mark.timeout(120)
def test_two_role_job_runs(tmp_path):
    tag = uuid.uuid4().hex[:8]
    job = (
        DLJobBuilder(f"j{tag}")
        .train(total=2).run(_ok_worker, tag)
        .role("evaluator").total(1).run(_ok_worker, tag)
        .build()
    )
    master = job.submit(blocking=True)
    assert master.status == "SUCCEEDED" and master.exit_code == 0
    for role, n in (("trainer", 2), ("evaluator", 1)):
        for r in range(n):
            assert os.path.exists(f"/tmp/dlrover_unified_{tag}_{role}_{r}")


@pytest.mark.timeout(120)
def test_role_failover(tmp_path):
    tag = uuid.uuid4().hex[:8]
    job = (
        DLJobBuilder(f"f{tag}")
        .train(total=2).run(_flaky_worker, tag).max_restarts(2)
        .build()
    )
    master = job.submit(blocking=True)
    assert master.status == "SUCCEEDED", master.status
    verts = master.graph.by_role("trainer")
    assert any(v.restarts > 0 for v in verts)
    st = PrimeMaster.load_state(master.state_path)
    assert st["status"] == "SUCCEEDED"


def _always_dies():
    raise SystemExit(5)


@pytest.mark.timeout(120)
def test_job_fails_after_budget(tmp_path):
    job = DLJobBuilder("dies").train(total=1).run(_always_dies).max_restarts(1).build()
    master = job.submit(blocking=True)
    assert master.status == "FAILED" and master.exit_code == 5


def test_scheduler_collocation_and_capacity():
    """Collocated roles' i-th workers share a bundle; bundles bin-pack into
    node capacity; overflow raises (ref schedule/scheduler.py:37-253)."""
    from dlrover_amd.unified.api import DLJobBuilder
    from dlrover_amd.unified.master import DLExecutionGraph
    from dlrover_amd.unified.scheduler import (
        NodeSpec,
        PlacementError,
        Scheduler,
    )

    def noop():
        pass

    job = (
        DLJobBuilder("sched")
        .role("actor").run(noop).total(4).resource(gpu=1)
        .role("rollout").run(noop).total(4).resource(gpu=1)
        .role("critic").run(noop).total(2).resource(gpu=2)
        .with_collocation("actor", "rollout")
        .build()
    )
    graph = DLExecutionGraph(job)
    pl = Scheduler([NodeSpec("n0", gpus=8), NodeSpec("n1", gpus=8)]).schedule(graph)
    # every actor-i and rollout-i landed on the SAME node
    for i in range(4):
        assert pl.assignments[f"actor-{i}"][0] == pl.assignments[f"rollout-{i}"][0]
    # total gpus per node within capacity
    used = {}
    for b in pl.bundles:
        used[b.node] = used.get(b.node, 0) + b.gpus
    assert all(v <= 8 for v in used.values()), used
    assert sum(used.values()) == 12

    # overflow: same job on a single 8-gpu node cannot fit
    try:
        Scheduler([NodeSpec("only", gpus=8)]).schedule(graph)
        raise AssertionError("expected PlacementError")
    except PlacementError:
        pass


def test_actor_rpc_roundtrip(tmp_path, monkeypatch):
    from dlrover_amd.unified.rpc import ActorRpcServer, call_actor

    monkeypatch.setenv("DLROVER_IPC_SOCKET_DIR", str(tmp_path))
    srv = ActorRpcServer("jobx", "trainer-0")
    srv.register("add", lambda a, b: a + b)
    srv.register("boom", lambda: 1 / 0)
    srv.start()
    try:
        assert call_actor("jobx", "trainer-0", "add", 2, 3) == 5
        assert call_actor("jobx", "trainer-0", "add", 2, b=4) == 6
        try:
            call_actor("jobx", "trainer-0", "boom")
            raise AssertionError("expected RuntimeError")
        except RuntimeError as e:
            assert "ZeroDivision" in str(e)
        try:
            call_actor("jobx", "trainer-0", "nope")
            raise AssertionError("expected RuntimeError")
        except RuntimeError as e:
            assert "no such method" in str(e)
    finally:
        srv.stop()


def test_prime_master_self_recovery(tmp_path):
    """A restarted master resumes per-vertex failover budgets from the
    state backend (ref manager.py:591-644)."""
    import json

    from dlrover_amd.unified.api import DLJobBuilder
    from dlrover_amd.unified.master import PrimeMaster

    def noop():
        pass

    state = tmp_path / "prime.json"
    state.write_text(json.dumps({
        "job": "rec", "status": "RUNNING",
        "vertices": [{"name": "train-0", "restarts": 2},
                     {"name": "train-1", "restarts": 1}],
    }))
    job = DLJobBuilder("rec").role("train").run(noop).total(2).build()
    m = PrimeMaster(job, state_path=str(state))
    budgets = {v.name: v.restarts for v in m.graph.vertices}
    assert budgets == {"train-0": 2, "train-1": 1}


def _noop():
    pass


def test_rl_builder_roles_and_validation(tmp_path):
    """RLJobBuilder (ref: api/builder/rl.py): role vocabulary + actor
    mandatory + unknown roles rejected; a valid RL graph runs end-to-end on
    the local backend."""
    import pytest

    from dlrover_amd.unified.api import RLJobBuilder

    with pytest.raises(ValueError, match="'actor' must be configured"):
        RLJobBuilder().trainer().run(_noop).build()

    with pytest.raises(ValueError, match="invalid role"):
        b = RLJobBuilder()
        b.actor().run(_noop)
        b.role("weird").run(_noop)
        b.build()

    job = (
        RLJobBuilder("rl-smoke")
        .trainer().run(_noop)
        .actor(2).run(_noop)
        .rollout(1).run(_noop)
        .with_collocation("actor", "rollout")
        .build()
    )
    assert set(job.roles) == {"trainer", "actor", "rollout"}
    assert job.roles["actor"].total == 2
    master = job.submit(blocking=True)
    assert master.status == "SUCCEEDED", master.status
