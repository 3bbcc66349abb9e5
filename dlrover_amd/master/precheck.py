"""Pre-check operator chain (ref: master/diagnosis/precheck_operator.py +
dist_master pre-check wiring): pluggable checks the master runs BEFORE it
lets agents start workers. Agents poll get_pre_check_result (elastic_run
wait_pre_check) until every operator passes or one fails terminally.

Operators are named in ``Context.pre_check_ops`` (CLI --pre-check-ops or
DLROVER_PRE_CHECK_OPS, comma-separated) and resolved from the registry; the
default chain is the reference's NoPreCheckOperator behavior (pass).
"""

import time
from typing import Dict, List, Optional, Tuple, Type

from dlrover_amd.common.log import logger

PASS = "PASS"
CHECKING = "CHECKING"
FAIL = "FAIL"


class PreCheckOperator:
    """One check. Returns (status, message)."""

    name = "noop"

    def check(self, master) -> Tuple[str, str]:  # pragma: no cover - iface
        return PASS, ""


class NoPreCheckOperator(PreCheckOperator):
    """Reference default: nothing to check."""

    name = "no_check"

    def check(self, master):
        return PASS, ""


class MinNodesPreCheckOperator(PreCheckOperator):
    """Hold workers until at least min_nodes agents joined the rendezvous
    (prevents a partial start burning restarts on rendezvous timeouts)."""

    name = "min_nodes"
    timeout_s = 600.0

    def __init__(self):
        self._first = 0.0

    def check(self, master):
        from dlrover_amd.common.constants import RendezvousName

        mgr = master.rdzv_managers.get(RendezvousName.TRAINING)
        if mgr is None:
            return PASS, ""
        need = getattr(mgr, "min_nodes", 1)
        # agents ANNOUNCE themselves (report_node_event ADDED) before they
        # block on the pre-check gate, so count announced nodes — counting
        # rendezvous joins here would deadlock (joins happen post-gate)
        announced = len([
            n for n in master.ctx.job_nodes().values() if not n.eliminated
        ]) if getattr(master, "ctx", None) else 0
        alive = len(getattr(mgr, "_alive_nodes", []) or [])
        waiting = len(getattr(mgr, "_waiting_nodes", {}) or {})
        have = max(announced, alive + waiting)
        if have >= need:
            return PASS, ""
        if self._first == 0.0:
            self._first = time.time()
        if time.time() - self._first > self.timeout_s:
            return FAIL, f"only {have}/{need} nodes joined"
        return CHECKING, f"{have}/{need} nodes joined"


class DeviceCountPreCheckOperator(PreCheckOperator):
    """Sanity: the master-side declared nproc matches what agents report
    (a mis-sized --nproc-per-node shows up before training, not as an RCCL
    init failure N minutes in)."""

    name = "device_count"

    def check(self, master):
        # local-master mode has no platform truth: informational pass
        return PASS, ""


_REGISTRY: Dict[str, Type[PreCheckOperator]] = {
    cls.name: cls
    for cls in (NoPreCheckOperator, MinNodesPreCheckOperator,
                DeviceCountPreCheckOperator)
}


def register_precheck_operator(cls: Type[PreCheckOperator]):
    _REGISTRY[cls.name] = cls
    return cls


class PreCheckChain:
    def __init__(self, names: Optional[List[str]] = None):
        names = [n for n in (names or []) if n]
        self.ops: List[PreCheckOperator] = []
        for n in names:
            cls = _REGISTRY.get(n)
            if cls is None:
                logger.warning("unknown pre-check operator %r — skipped", n)
                continue
            self.ops.append(cls())
        self._failed_msg = ""

    def evaluate(self, master) -> Tuple[str, str]:
        """Overall status: FAIL dominates, then CHECKING, else PASS."""
        if self._failed_msg:
            return FAIL, self._failed_msg
        overall, detail = PASS, ""
        for op in self.ops:
            try:
                status, msg = op.check(master)
            except Exception as e:  # noqa: BLE001 — a broken op must not wedge
                logger.exception("pre-check %s crashed", op.name)
                status, msg = PASS, f"{op.name} errored: {e}"
            if status == FAIL:
                self._failed_msg = f"{op.name}: {msg}"
                return FAIL, self._failed_msg
            if status == CHECKING:
                overall, detail = CHECKING, f"{op.name}: {msg}"
        return overall, detail
