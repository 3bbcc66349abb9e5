"""Node health / network check agent.

Parity target: ref training.py:2055-2445 (NodeCheckElasticAgent: two probe
rounds on the NETWORK_CHECK rendezvous plane, report status+elapsed, query
fault/straggler, raise NodeCheckFailedError when this node is judged bad) and
trainer/torch/node_check/* probe mains.

Round 0 pairs adjacent nodes; round 1 re-pairs fastest<->slowest (master-side
NetworkCheckRendezvousManager) so a bad node/link is isolated in 2 rounds.
"""

import time
from typing import Optional

from torch.distributed.elastic.agent.server.api import WorkerSpec
from torch.distributed.elastic.agent.server.local_elastic_agent import (
    LocalElasticAgent,
)
from torch.distributed.elastic.multiprocessing import DefaultLogsSpecs

from dlrover_amd.agent.master_client import MasterClient
from dlrover_amd.agent.training import ElasticLaunchConfig, MasterRendezvousHandler
from dlrover_amd.common.constants import RendezvousName
from dlrover_amd.common.log import logger


class NodeCheckFailedError(RuntimeError):
    pass


CHECK_ROUNDS = 2


def _run_one_round(config: ElasticLaunchConfig, client: MasterClient) -> float:
    """Run the probe workers once through a NETWORK_CHECK rendezvous; returns
    elapsed seconds (inf on failure)."""
    handler = MasterRendezvousHandler(
        RendezvousName.NETWORK_CHECK,
        config.node_rank,
        config.nproc_per_node,
        client=client,
        rdzv_timeout=max(config.rdzv_timeout, 120),
    )
    spec = WorkerSpec(
        role="node-check",
        local_world_size=config.nproc_per_node,
        entrypoint="python",
        args=("-m", "dlrover_amd.trainer.node_check.amd_gpu"),
        rdzv_handler=handler,
        max_restarts=0,
        monitor_interval=1.0,
    )
    agent = LocalElasticAgent(spec, logs_specs=DefaultLogsSpecs(log_dir=None))
    start = time.time()
    try:
        result = agent.run()
        elapsed = time.time() - start
        if result.is_failed():
            logger.warning("node check workers failed: %s", result.failures)
            return float("inf")
        return elapsed
    except Exception:  # noqa: BLE001
        logger.exception("node check round crashed")
        return float("inf")
    finally:
        handler.shutdown()


def run_network_check(
    config: ElasticLaunchConfig, client: Optional[MasterClient] = None
) -> bool:
    """ref: node_health_check / run_network_check (training.py:2316-2445)."""
    client = client or MasterClient.singleton_instance()
    for rnd in range(CHECK_ROUNDS):
        elapsed = _run_one_round(config, client)
        normal = elapsed != float("inf")
        client.report_network_check_result(
            config.node_rank, normal, elapsed if normal else 3600.0
        )
        logger.info(
            "node check round %s: normal=%s elapsed=%.2fs", rnd, normal, elapsed
        )
        # wait for the master verdict over this round before the next
        deadline = time.time() + 300
        while time.time() < deadline:
            fault_nodes, reason = client.check_fault_node()
            if reason != "not_initialized":
                break
            time.sleep(2)
        if config.node_rank in fault_nodes and rnd == CHECK_ROUNDS - 1:
            raise NodeCheckFailedError(
                f"node {config.node_rank} failed the network check"
            )
    stragglers = client.check_straggler()
    if config.node_rank in stragglers:
        if getattr(config, "exclude_straggler", False):
            # ref: --exclude-straggler (elastic_run.py:170) — the agent
            # refuses to train on a straggler node so the platform
            # reschedules it; default is observe-and-continue
            raise NodeCheckFailedError(
                f"node {config.node_rank} is a straggler and "
                "--exclude-straggler is set"
            )
        logger.warning("this node is a straggler (continuing; master decides)")
    fault_nodes, _ = client.check_fault_node()
    if config.node_rank in fault_nodes:
        raise NodeCheckFailedError(f"node {config.node_rank} is a fault node")
    return True
