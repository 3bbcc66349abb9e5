"""Master-side node lifecycle management.

Parity target: ref dlrover/python/master/node/dist_job_manager.py:107-1664
(node monitoring, heartbeat dead-node detection :604-682, status-flow event
processing :862, relaunch ladder :1083-1224) and local_job_manager.py:25-174.

Split:
  - JobManager (base): node registry, heartbeats, events, relaunch decisions,
    diagnosis-action delivery — everything platform-independent;
  - LocalJobManager: standalone mode — the "cluster" is the local agents that
    joined; relaunch = push a RESTART_WORKER action to the agent;
  - DistributedJobManager: adds a Scaler (create/replace nodes) and a
    NodeWatcher (platform events); k8s implementations live in
    dlrover_amd.master.scaler / watcher, simulations in dlrover_amd.testing.
"""

import threading
import time
from typing import List, Optional, Tuple

from dlrover_amd.common import comm
from dlrover_amd.common.constants import (
    JobExitReason,
    NodeEventType,
    NodeExitReason,
    NodeStatus,
    NodeType,
    TrainingExceptionLevel,
)
from dlrover_amd.common.global_context import Context
from dlrover_amd.common.log import logger
from dlrover_amd.common.node import Node
from dlrover_amd.diagnosis.actions import (
    DiagnosisActionType,
    JobAbortAction,
    NodeAction,
)
from dlrover_amd.master.node.job_context import JobContext
from dlrover_amd.master.node.status_flow import allowed_transition


class JobManager:
    def __init__(self, job_context: Optional[JobContext] = None, rdzv_managers=None):
        self.ctx = job_context or JobContext.singleton_instance()
        self.rdzv_managers = rdzv_managers or {}
        self._config = Context.singleton_instance()
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._failures: List[comm.NodeFailure] = []
        # pluggable node-event callbacks (ref: event_callback.py —
        # TaskRescheduleCallback re-shards a dead worker's data,
        # AllReduceNodeHandlingCallback drives rdzv):
        # each is called as cb(node, reason) when a node FAILS
        self.node_failure_callbacks: List = []

    def add_node_failure_callback(self, cb):
        self.node_failure_callbacks.append(cb)

    # -- lifecycle -----------------------------------------------------------------

    def start(self):
        t = threading.Thread(
            target=self._heartbeat_monitor, name="hb-monitor", daemon=True
        )
        t.start()
        self._threads.append(t)

    def stop(self):
        self._stop.set()
        for t in self._threads:
            t.join(timeout=3)

    # -- node registry ----------------------------------------------------------------

    def on_node_joined(self, node_rank: int, node_ip: str = ""):
        node = self.ctx.get_node(NodeType.WORKER, node_rank)
        if node is None:
            node = Node(
                NodeType.WORKER,
                node_rank,
                rank_index=node_rank,
                max_relaunch_count=self._config.relaunch_error_max,
            )
        node.host_ip = node_ip
        node.update_status(NodeStatus.RUNNING)
        node.update_heartbeat()
        self.ctx.update_node(node)
        for mgr in self.rdzv_managers.values():
            mgr.add_alive_node(node_rank)

    def on_heartbeat(
        self, node_id: int, node_rank: int, ts: float
    ) -> Optional[Tuple[str, dict]]:
        node = self.ctx.get_node(NodeType.WORKER, node_id)
        if node is not None:
            node.update_heartbeat(ts)
        action = self.ctx.next_action(node_id)
        return action.to_wire() if action is not None else None

    def running_nodes(self) -> List[comm.NodeMeta]:
        metas = []
        for node in self.ctx.alive_nodes():
            metas.append(
                comm.NodeMeta(
                    type=node.type,
                    id=node.id,
                    rank=node.rank_index,
                    addr=node.host_ip,
                    status=node.status,
                )
            )
        return metas

    def training_status(self) -> str:
        return self.ctx.job_stage

    # -- events -----------------------------------------------------------------------

    def on_node_event(self, event: comm.NodeEvent):
        meta = event.node
        if meta is None:
            return
        node = self.ctx.get_node(NodeType.WORKER, meta.id)
        if node is None:
            node = Node(NodeType.WORKER, meta.id, rank_index=meta.rank)
            self.ctx.update_node(node)
        if event.event_type == NodeEventType.SUCCEEDED_EXITED:
            self._transition(node, NodeStatus.SUCCEEDED)
            self._maybe_finish_job()
        elif event.event_type == NodeEventType.FAILED_EXITED:
            if event.reason == NodeExitReason.OOM:
                node.exit_reason = NodeExitReason.OOM
            node.exit_reason = node.exit_reason or NodeExitReason.UNKNOWN_ERROR
            self._transition(node, NodeStatus.FAILED)
            self._handle_node_failure(node, event.reason)
        elif event.event_type == NodeEventType.NODE_CHECK_FAILED:
            node.eliminated = True
            self._transition(node, NodeStatus.BREAKDOWN)
            self._handle_node_failure(node, "node check failed")
        elif event.event_type == NodeEventType.DELETED:
            self._transition(node, NodeStatus.DELETED)
            self._handle_node_failure(node, "node deleted")
        elif event.event_type in (NodeEventType.ADDED, NodeEventType.MODIFIED):
            if meta.status and allowed_transition(node.status, meta.status):
                node.update_status(meta.status)
                if meta.status == NodeStatus.FAILED:
                    if event.reason == NodeExitReason.OOM:
                        node.exit_reason = NodeExitReason.OOM
                    self._handle_node_failure(node, event.reason)

    def _transition(self, node: Node, status: str):
        if allowed_transition(node.status, status):
            node.update_status(status)
            self.ctx.update_node(node)
        else:
            logger.info(
                "ignored status transition %s -> %s for %s", node.status, status, node
            )

    def on_node_failure(self, msg: comm.NodeFailure):
        self._failures.append(msg)
        node = self.ctx.get_node(NodeType.WORKER, msg.node_id)
        if node is not None and msg.level == TrainingExceptionLevel.NODE_ERROR:
            node.exit_reason = NodeExitReason.HARDWARE_ERROR

    # -- failure handling / relaunch ladder (ref: _should_relaunch :1083) ---------------

    def _handle_node_failure(self, node: Node, reason: str = ""):
        from dlrover_amd.common.events import master_events

        master_events().instant(
            "node_fail",
            {"node": node.id, "rank": node.rank_index,
             "exit_reason": node.exit_reason, "reason": reason,
             "relaunches": node.relaunch_count},
        )
        for mgr in self.rdzv_managers.values():
            mgr.remove_alive_node(node.id)
        for cb in self.node_failure_callbacks:
            try:
                cb(node, reason)
            except Exception:  # noqa: BLE001 — callbacks must not block recovery
                logger.exception("node-failure callback %s failed", cb)
        if node.exit_reason == NodeExitReason.NO_HEARTBEAT:
            # the node's AGENT is gone (scale-down, preemption, host death):
            # nobody is listening for a restart action. Shrink the job — the
            # survivors re-rendezvous; abort only if nobody is left.
            node.relaunchable = False
            self.ctx.update_node(node)
            if not self.ctx.alive_nodes():
                logger.error("all nodes gone: stopping job")
                self.ctx.request_stop(JobExitReason.WORKER_ERROR, code=1)
            return
        if node.is_unrecoverable_failure():
            others = [n for n in self.ctx.alive_nodes() if n.id != node.id]
            if others:
                # eliminate just this node; the rest of the job continues
                logger.error(
                    "node %s unrecoverable (%s): eliminating it, job continues "
                    "with %s nodes", node, reason, len(others),
                )
                node.eliminated = True
                node.relaunchable = False
                self.ctx.update_node(node)
                return
            logger.error("node %s unrecoverable (%s): aborting job", node, reason)
            self.ctx.enqueue_action(
                JobAbortAction(node_id=-1, reason=f"node {node.id}: {reason}")
            )
            self.ctx.request_stop(JobExitReason.WORKER_ERROR, code=1)
            return
        if node.should_relaunch():
            node.inc_relaunch_count()
            self._relaunch_node(node, reason)

    def _relaunch_node(self, node: Node, reason: str):
        raise NotImplementedError

    def _maybe_finish_job(self):
        nodes = self.ctx.job_nodes()
        if nodes and all(n.status == NodeStatus.SUCCEEDED for n in nodes.values()):
            self.ctx.request_stop(JobExitReason.SUCCEEDED, code=0)

    # -- heartbeat monitor (ref: dist_job_manager.py:604-682) ----------------------------

    def _heartbeat_monitor(self):
        timeout = self._config.heartbeat_timeout
        while not self._stop.wait(5.0):
            now = time.time()
            for node in self.ctx.alive_nodes():
                if node.heartbeat_time <= 0 or node.status != NodeStatus.RUNNING:
                    continue
                if now - node.heartbeat_time > timeout:
                    logger.warning(
                        "node %s missed heartbeats for %.0fs — marking failed",
                        node,
                        now - node.heartbeat_time,
                    )
                    node.exit_reason = NodeExitReason.NO_HEARTBEAT
                    self._transition(node, NodeStatus.FAILED)
                    self._handle_node_failure(node, "heartbeat timeout")


class LocalJobManager(JobManager):
    """Standalone mode (ref: local_job_manager.py): the node IS the local
    agent; relaunching a 'pod' is impossible, so failures become
    RESTART_WORKER actions delivered on the agent's next heartbeat."""

    def _relaunch_node(self, node: Node, reason: str):
        node.update_status(NodeStatus.RUNNING)  # same process carries on
        self.ctx.update_node(node)
        self.ctx.enqueue_action(
            NodeAction(
                action_type=DiagnosisActionType.RESTART_WORKER,
                node_id=node.id,
                reason=reason or "node failure",
            )
        )


class DistributedJobManager(JobManager):
    """Cluster mode: relaunch via the platform scaler and watch platform
    events (ref: dist_job_manager.py)."""

    def __init__(self, scaler=None, watcher=None, min_nodes: int = 1,
                 job_watcher=None, scaleplan_watcher=None, **kw):
        super().__init__(**kw)
        self.scaler = scaler
        self.watcher = watcher
        self.job_watcher = job_watcher  # ElasticJob CR spec edits
        self.scaleplan_watcher = scaleplan_watcher  # manual ScalePlan CRs
        self.min_nodes = min_nodes
        self._next_node_id = 1000  # relaunched nodes get fresh ids
        self._pending_since: dict = {}  # node id -> first PENDING sighting
        self._group_failures: dict = {}  # group id -> [fail timestamps]
        self._auto_scaler = None

    def start(self):
        super().start()
        if self.watcher is not None:
            t = threading.Thread(
                target=self._watch_events, name="node-watcher", daemon=True
            )
            t.start()
            self._threads.append(t)
        if self.job_watcher is not None:
            t = threading.Thread(
                target=self._watch_job_spec, name="job-watcher", daemon=True
            )
            t.start()
            self._threads.append(t)
        if self.scaleplan_watcher is not None:
            t = threading.Thread(
                target=self._watch_scaleplans, name="scaleplan-watcher",
                daemon=True,
            )
            t.start()
            self._threads.append(t)
        t = threading.Thread(
            target=self._pending_monitor, name="pending-monitor", daemon=True
        )
        t.start()
        self._threads.append(t)
        # periodic optimizer->scaler loop (ref: dist_job_manager starting
        # AllreduceTrainingAutoScaler), opt-in via Context.auto_worker_enabled
        if self.scaler is not None and self._config.auto_worker_enabled:
            from dlrover_amd.master.auto_scale import (
                JobAutoScaler,
                LocalResourceOptimizer,
            )

            perf = getattr(self, "perf_monitor", None)
            opt = LocalResourceOptimizer(
                perf, ctx=self.ctx, min_nodes=self.min_nodes,
                max_nodes=max(self.min_nodes,
                              getattr(self, "max_nodes", self.min_nodes)),
            )
            self._auto_scaler = JobAutoScaler(opt, self.scaler).start()

    def stop(self):
        if self._auto_scaler is not None:
            self._auto_scaler.stop()
        super().stop()

    # -- early stop on unschedulable nodes (ref: dist_job_manager.py:386
    # -- _early_stop_part_of_pending) -----------------------------------------

    def _pending_monitor(self):
        """A job whose nodes sit PENDING past seconds_to_wait_pending can
        never reach min_nodes: stop early and release what it holds rather
        than occupying partial resources forever."""
        wait = self._config.seconds_to_wait_pending
        while not self._stop.wait(5.0):
            now = time.time()
            pending = [
                n for n in self.ctx.job_nodes().values()
                if n.status in (NodeStatus.PENDING, NodeStatus.INITIAL)
                and not n.eliminated
            ]
            seen = {n.id for n in pending}
            for nid in list(self._pending_since):
                if nid not in seen:
                    del self._pending_since[nid]
            overdue = []
            for n in pending:
                first = self._pending_since.setdefault(n.id, now)
                if now - first > wait:
                    overdue.append(n)
            if not overdue:
                continue
            running = [
                n for n in self.ctx.alive_nodes()
                if n.status == NodeStatus.RUNNING
            ]
            if len(running) >= self.min_nodes:
                # enough schedulable capacity: give up on the stragglers
                for n in overdue:
                    logger.warning(
                        "node %s pending > %.0fs: eliminating (job keeps "
                        "%s running nodes)", n, wait, len(running),
                    )
                    n.eliminated = True
                    n.relaunchable = False
                    self.ctx.update_node(n)
                    del self._pending_since[n.id]
                continue
            logger.error(
                "%s nodes pending > %.0fs and only %s/%s running: "
                "early-stopping the job (unschedulable)",
                len(overdue), wait, len(running), self.min_nodes,
            )
            self.ctx.enqueue_action(
                JobAbortAction(node_id=-1, reason="pending timeout")
            )
            self.ctx.request_stop(JobExitReason.PENDING_TIMEOUT, code=1)
            return

    def _watch_events(self):
        while not self._stop.is_set():
            try:
                for event in self.watcher.watch():
                    if self._stop.is_set():
                        return
                    self.on_node_event(event)
            except Exception:  # noqa: BLE001 — watch streams break routinely
                logger.exception("node watcher stream broke; re-watching")
                time.sleep(3)

    def _watch_scaleplans(self):
        """Apply user-submitted manual ScalePlan CRs (ref:
        K8sScalePlanWatcher -> JobAutoScaler execution)."""
        while not self._stop.is_set():
            try:
                for plan in self.scaleplan_watcher.watch():
                    if self._stop.is_set():
                        return
                    logger.info("manual ScalePlan: scale to %s (%s)",
                                plan.node_count, plan.comment)
                    if self.scaler is not None and plan.node_count > 0:
                        self.scaler.scale_to(
                            plan.node_count,
                            list(self.ctx.job_nodes().values()),
                        )
            except Exception:  # noqa: BLE001 — watch streams break routinely
                logger.exception("scaleplan watcher stream broke; re-watching")
                time.sleep(3)

    def _watch_job_spec(self):
        """Consume ElasticJob CR edits (ref: K8sElasticJobWatcher): replica
        edits scale the worker group; suspend scales to 0 and resume
        restores the pre-suspend size."""
        pre_suspend = 0
        while not self._stop.is_set():
            try:
                for kind, plan in self.job_watcher.watch():
                    if self._stop.is_set():
                        return
                    nodes = list(self.ctx.job_nodes().values())
                    alive = [n for n in nodes if n.is_alive()]
                    if kind == "scale" and plan is not None:
                        logger.info("ElasticJob spec: scale to %s (%s)",
                                    plan.node_count, plan.comment)
                        if self.scaler is not None:
                            self.scaler.scale_to(plan.node_count, nodes)
                    elif kind == "suspend":
                        pre_suspend = len(alive)
                        logger.warning(
                            "ElasticJob suspended: releasing %s workers",
                            pre_suspend,
                        )
                        if self.scaler is not None:
                            self.scaler.scale_to(0, nodes)
                    elif kind == "resume" and pre_suspend > 0:
                        logger.info("ElasticJob resumed: scaling back to %s",
                                    pre_suspend)
                        if self.scaler is not None:
                            self.scaler.scale_to(pre_suspend, nodes)
                        pre_suspend = 0
            except Exception:  # noqa: BLE001 — watch streams break routinely
                logger.exception("job watcher stream broke; re-watching")
                time.sleep(3)

    def _relaunch_node(self, node: Node, reason: str):
        if self.scaler is None:
            # no platform scaler: degrade to local behavior
            self.ctx.enqueue_action(
                NodeAction(
                    action_type=DiagnosisActionType.RELAUNCH_WORKER,
                    node_id=node.id,
                    reason=reason,
                )
            )
            return
        if self._maybe_group_relaunch(node, reason):
            return
        self._relaunch_one(node, reason)

    OOM_MEMORY_FACTOR = 2.0  # ref: PSTrainingAutoScaler OOM recovery cushion
    OOM_MEMORY_CAP_MB = 1 << 20  # 1 TiB

    def _relaunch_one(self, node: Node, reason: str) -> Node:
        replacement = node.new_incarnation(self._next_node_id)
        if node.exit_reason == NodeExitReason.OOM and (
            replacement.config_resource.memory_mb > 0
        ):
            grown = int(
                min(
                    replacement.config_resource.memory_mb
                    * self.OOM_MEMORY_FACTOR,
                    self.OOM_MEMORY_CAP_MB,
                )
            )
            logger.info(
                "node %s OOMKilled: relaunching with memory %s -> %s MB",
                node.id, replacement.config_resource.memory_mb, grown,
            )
            replacement.config_resource.memory_mb = grown
        self._next_node_id += 1
        self.ctx.update_node(replacement)
        logger.info("relaunching %s as %s (%s)", node, replacement, reason)
        from dlrover_amd.common.events import master_events

        master_events().instant(
            "node_relaunch",
            {"node": node.id, "replacement": replacement.id,
             "rank": node.rank_index, "reason": reason,
             "memory_mb": replacement.config_resource.memory_mb},
        )
        self.scaler.launch_node(replacement)
        self.scaler.remove_node(node)
        return replacement

    # -- node-group relaunch (ref: dist_job_manager.py:1224) -------------------

    GROUP_FAIL_WINDOW = 300.0  # s
    GROUP_FAIL_THRESHOLD = 2

    def _maybe_group_relaunch(self, node: Node, reason: str) -> bool:
        """Multiple failures inside one node GROUP (super-pod / switch
        domain) within a window indicate fabric-level trouble: relaunch the
        WHOLE group together so the replacement set can be placed on a
        healthy domain, instead of trickling single pods back into the bad
        one."""
        if node.group is None:
            return False
        now = time.time()
        hist = self._group_failures.setdefault(node.group, [])
        hist.append(now)
        hist[:] = [t for t in hist if now - t < self.GROUP_FAIL_WINDOW]
        if len(hist) < self.GROUP_FAIL_THRESHOLD:
            return False
        peers = [
            n for n in self.ctx.job_nodes().values()
            if n.group == node.group and not n.eliminated
            and n.id != node.id
            and n.status in (NodeStatus.RUNNING, NodeStatus.PENDING,
                             NodeStatus.FAILED)
        ]
        logger.warning(
            "group %s: %s failures in %.0fs — relaunching the whole group "
            "(%s peers + the failed node)",
            node.group, len(hist), self.GROUP_FAIL_WINDOW, len(peers),
        )
        self._group_failures[node.group] = []
        self._relaunch_one(node, f"group relaunch: {reason}")
        for peer in peers:
            if peer.should_relaunch():
                peer.inc_relaunch_count()
                self._relaunch_one(peer, f"group relaunch ({node.group})")
        return True
