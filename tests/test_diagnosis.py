"""Failure classification + hang diagnostician behaviors."""

import json
import time

import pytest

from dlrover_amd.common import comm
from dlrover_amd.common.global_context import Context
from dlrover_amd.diagnosis.diagnosis_agent import (
    ABORT_JOB,
    RELAUNCH_NODE,
    RESTART_WORKER,
    TrainingLogCollector,
    WorkerDiagnosisAgent,
    classify_error,
)
from dlrover_amd.master.diagnosis_master import DiagnosisMaster
from dlrover_amd.master.node.job_context import JobContext
from dlrover_amd.master.perf_monitor import PerfMonitor


def test_classify_software_error():
    assert classify_error("RuntimeError: shape mismatch [2,3] vs [4]") == RESTART_WORKER


def test_classify_hardware_error():
    assert classify_error("Memory access fault by GPU node-2") == RELAUNCH_NODE
    assert classify_error("RCCL WARN unhandled system error, NCCL version ...") == (
        RELAUNCH_NODE
    )
    assert classify_error("HSA_STATUS_ERROR_OUT_OF_RESOURCES") == RELAUNCH_NODE


def test_classify_fatal_user_error():
    assert classify_error("ModuleNotFoundError: No module named 'foo'") == ABORT_JOB
    # OOM relaunches (the master grows the replacement pod's memory);
    # repeated OOMs exhaust the budget and only then abort
    assert classify_error("torch.OutOfMemoryError: HIP out of memory") == RELAUNCH_NODE


def test_restart_budget_escalates():
    agent = WorkerDiagnosisAgent()
    assert agent.diagnose_training_failure("ValueError: x", 0, 3) == RESTART_WORKER
    assert agent.diagnose_training_failure("ValueError: x", 3, 3) == RELAUNCH_NODE


def test_log_collector(tmp_path):
    p = tmp_path / "worker.log"
    p.write_text("\n".join(f"line{i}" for i in range(500)))
    log = TrainingLogCollector(str(p), n_lines=100).collect()
    assert len(log.lines) == 100 and log.lines[-1] == "line499"


def test_hang_diagnostician_metric_path(monkeypatch):
    JobContext._reset_for_tests()
    Context._reset_for_tests()
    ctx = JobContext.singleton_instance()
    cfg = Context.singleton_instance()
    cfg.hang_downtime = 1  # 1s for the test
    perf = PerfMonitor()
    dm = DiagnosisMaster(perf, ctx)
    old = time.time() - 10
    for nid in (0, 1):
        dm.collect_data(
            comm.DiagnosisReportData(
                data_cls="hang",
                data_content=json.dumps({"hang": True, "since": old}),
                node_id=nid,
            )
        )
    dm._check_hang()
    action = ctx.next_action(-1)
    assert action is not None and action.action_type == "restart_worker"
    # one rank NOT hung -> no further action
    dm._last_hang_action = 0
    dm.collect_data(
        comm.DiagnosisReportData(
            data_cls="hang",
            data_content=json.dumps({"hang": False, "since": 0}),
            node_id=1,
        )
    )
    dm._check_hang()
    assert ctx.next_action(-1) is None
    JobContext._reset_for_tests()
    Context._reset_for_tests()


def test_hang_diagnostician_step_path():
    JobContext._reset_for_tests()
    Context._reset_for_tests()
    ctx = JobContext.singleton_instance()
    Context.singleton_instance().hang_downtime = 1
    perf = PerfMonitor()
    dm = DiagnosisMaster(perf, ctx)
    perf.report_global_step(10, time.time() - 30)  # stale progress
    dm._check_hang()
    action = ctx.next_action(-1)
    assert action is not None and "hang" in action.reason
    JobContext._reset_for_tests()
    Context._reset_for_tests()


def test_precheck_operator_chain(monkeypatch):
    """Pluggable pre-check chain gates worker start (ref
    precheck_operator.py): min_nodes holds CHECKING, fails terminally on
    timeout, and the master serves the aggregated status."""
    from dlrover_amd.master.precheck import (
        CHECKING,
        FAIL,
        PASS,
        MinNodesPreCheckOperator,
        PreCheckChain,
        PreCheckOperator,
        register_precheck_operator,
    )

    class Master:
        rdzv_managers = {}

    chain = PreCheckChain(["no_check", "bogus_name"])
    assert chain.evaluate(Master()) == (PASS, "")

    class Rdzv:
        _alive_nodes = [0]
        _waiting_nodes = {}
        min_nodes = 2

    from dlrover_amd.common.constants import RendezvousName

    m = Master()
    m.rdzv_managers = {RendezvousName.TRAINING: Rdzv()}
    chain = PreCheckChain(["min_nodes"])
    status, msg = chain.evaluate(m)
    assert status == CHECKING and "1/2" in msg
    # timeout -> terminal FAIL (sticky)
    chain.ops[0].timeout_s = 0.0
    status, _ = chain.evaluate(m)
    assert status == FAIL
    Rdzv._waiting_nodes = {1: 1}
    status, _ = chain.evaluate(m)
    assert status == FAIL  # sticky once failed

    # satisfied chain passes
    chain2 = PreCheckChain(["min_nodes"])
    Rdzv._alive_nodes = [0, 1]
    assert chain2.evaluate(m)[0] == PASS

    # custom registered operator
    @register_precheck_operator
    class Always(PreCheckOperator):
        name = "always_fail"

        def check(self, master):
            return FAIL, "nope"

    chain3 = PreCheckChain(["always_fail"])
    assert chain3.evaluate(m)[0] == FAIL


def test_failover_extension_overrides_ladder(tmp_path, monkeypatch):
    """DLROVER_FAILOVER_EXTENSION names a user class whose strategy overrides
    the built-in ladder (ref: torch/dynamic_failover.py); invalid verdicts
    and load failures fall back to the ladder."""
    import sys

    ext = tmp_path / "my_failover.py"
    ext.write_text(
        "class AlwaysRelaunch:\n"
        "    def get_user_failover_strategy(self, text, restarts):\n"
        "        return 'relaunch' if 'special' in text else None\n"
    )
    monkeypatch.syspath_prepend(str(tmp_path))
    monkeypatch.setenv("DLROVER_FAILOVER_EXTENSION", "my_failover:AlwaysRelaunch")
    sys.modules.pop("my_failover", None)

    from dlrover_amd.diagnosis.diagnosis_agent import (
        RELAUNCH_NODE,
        RESTART_WORKER,
        WorkerDiagnosisAgent,
    )

    agent = WorkerDiagnosisAgent()
    assert agent.extension is not None
    # extension claims this one
    assert agent.diagnose_training_failure("a special failure", 0, 3) == RELAUNCH_NODE
    # extension defers (returns None) -> ladder says restart
    assert agent.diagnose_training_failure("ordinary traceback", 0, 3) == RESTART_WORKER

    # broken spec -> no extension, ladder still works
    monkeypatch.setenv("DLROVER_FAILOVER_EXTENSION", "nope.not.there:X")
    agent2 = WorkerDiagnosisAgent()
    assert agent2.extension is None
    assert agent2.diagnose_training_failure("ordinary", 0, 3) == RESTART_WORKER
