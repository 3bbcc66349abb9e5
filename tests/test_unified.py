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
