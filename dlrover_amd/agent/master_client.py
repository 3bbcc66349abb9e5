"""Agent/worker-side client for the master RPC service.

Parity target: ref dlrover/python/elastic_agent/master_client.py:51-600 —
one method per servicer route, retry decorator, singleton bound to
DLROVER_MASTER_ADDR. Transport matches the master's --service_type.
"""

import functools
import os
import threading
import time
from typing import Dict, List, Optional, Tuple

from dlrover_amd.common import comm
from dlrover_amd.common.comm import BaseRequest, BaseResponse
from dlrover_amd.common.constants import CommServiceType, NodeEnv, RendezvousName
from dlrover_amd.common.log import logger
from dlrover_amd.utils.transport import create_rpc_client


def retry_rpc(retries: int = 3, delay: float = 1.0):
    def deco(fn):
        @functools.wraps(fn)
        def wrapped(*args, **kwargs):
            err = None
            for i in range(retries):
                try:
                    return fn(*args, **kwargs)
                except Exception as e:  # noqa: BLE001
                    err = e
                    logger.warning(
                        "RPC %s attempt %s/%s failed: %s", fn.__name__, i + 1, retries, e
                    )
                    time.sleep(delay * (i + 1))
            raise err

        return wrapped

    return deco


class MasterClient:
    _instance: Optional["MasterClient"] = None
    _lock = threading.Lock()

    def __init__(self, master_addr: str, node_id: int = 0, node_type: str = "worker",
                 service_type: str = CommServiceType.TCP):
        self.master_addr = master_addr
        self.node_id = node_id
        self.node_type = node_type
        self._client = create_rpc_client(service_type, master_addr)

    # -- singleton ---------------------------------------------------------------

    @classmethod
    def singleton_instance(cls) -> "MasterClient":
        if cls._instance is None:
            with cls._lock:
                if cls._instance is None:
                    addr = os.getenv(NodeEnv.MASTER_ADDR, "")
                    if not addr:
                        raise RuntimeError(
                            f"{NodeEnv.MASTER_ADDR} is not set — is this process "
                            "running under dlrover-run?"
                        )
                    cls._instance = cls(
                        addr,
                        node_id=int(os.getenv(NodeEnv.NODE_ID, "0")),
                        node_type=os.getenv("NODE_TYPE", "worker"),
                        service_type=os.getenv(
                            NodeEnv.MASTER_SERVICE_TYPE, CommServiceType.TCP
                        ),
                    )
        return cls._instance

    @classmethod
    def reset(cls):
        with cls._lock:
            if cls._instance is not None:
                cls._instance.close()
            cls._instance = None

    def close(self):
        self._client.close()

    # -- raw verbs ---------------------------------------------------------------

    def _req(self, data) -> BaseRequest:
        return BaseRequest(node_id=self.node_id, node_type=self.node_type, data=data)

    @retry_rpc()
    def get(self, msg) -> Optional[comm.Message]:
        resp: BaseResponse = self._client.call("get", self._req(msg))
        if not resp.success:
            raise RuntimeError(f"master get failed: {resp.reason}")
        return resp.data

    @retry_rpc()
    def report(self, msg) -> Optional[comm.Message]:
        resp: BaseResponse = self._client.call("report", self._req(msg))
        if not resp.success:
            raise RuntimeError(f"master report failed: {resp.reason}")
        return resp.data

    # -- rendezvous ---------------------------------------------------------------

    def join_rendezvous(
        self, node_rank: int, local_world_size: int,
        rdzv_name: str = RendezvousName.TRAINING, node_ip: str = "",
    ) -> int:
        resp = self.report(
            comm.JoinRendezvousRequest(
                node_id=self.node_id,
                node_rank=node_rank,
                local_world_size=local_world_size,
                rdzv_name=rdzv_name,
                node_ip=node_ip,
            )
        )
        return resp.round if resp else 0

    def get_comm_world(
        self, rdzv_name: str, node_rank: int
    ) -> Tuple[int, int, Dict[int, int]]:
        resp = self.get(comm.CommWorldRequest(node_id=node_rank, rdzv_name=rdzv_name))
        return resp.rdzv_round, resp.group, resp.world

    def num_nodes_waiting(self, rdzv_name: str = RendezvousName.TRAINING) -> int:
        resp = self.get(comm.WaitingNodeNumRequest(rdzv_name=rdzv_name))
        return resp.waiting_num

    def block_rendezvous(
        self, node_rank: int, blocked: bool,
        rdzv_name: str = RendezvousName.TRAINING,
    ):
        """Hold/release the pending round while this node persists UCP
        shards (ref: UcpRdzvManager blockable rendezvous)."""
        self.report(
            comm.RdzvBlockRequest(
                node_rank=node_rank, blocked=blocked, rdzv_name=rdzv_name
            )
        )

    def report_rdzv_params(
        self, min_nodes: int, max_nodes: int, waiting_timeout: float, node_unit: int
    ):
        self.report(
            comm.RendezvousParams(
                min_nodes=min_nodes,
                max_nodes=max_nodes,
                waiting_timeout=waiting_timeout,
                node_unit=node_unit,
            )
        )

    # -- kv store -------------------------------------------------------------------

    def kv_store_get(self, key: str) -> bytes:
        resp = self.get(comm.KVStoreGetRequest(key=key))
        return resp.value

    def kv_store_multi_get(self, keys: List[str]) -> Dict[str, bytes]:
        resp = self.get(comm.KVStoreMultiGetRequest(keys=keys))
        return resp.kvs

    def kv_store_set(self, key: str, value: bytes):
        self.report(comm.KeyValuePair(key=key, value=value))

    def kv_store_multi_set(self, kvs: Dict[str, bytes]):
        self.report(comm.KeyValuePairs(kvs=kvs))

    def kv_store_add(self, key: str, amount: int) -> int:
        resp = self.get(comm.KVStoreAddRequest(key=key, amount=amount))
        return resp.value

    def kv_store_delete(self, key: str):
        self.report(comm.KVStoreDeleteRequest(key=key))

    # -- health / lifecycle ------------------------------------------------------------

    def report_heart_beat(self, node_rank: int = -1) -> comm.HeartbeatResponse:
        return self.report(
            comm.HeartbeatRequest(
                node_id=self.node_id, node_rank=node_rank, timestamp=time.time()
            )
        )

    def report_node_event(self, event_type: str, reason: str = "", rank: int = -1):
        meta = comm.NodeMeta(type=self.node_type, id=self.node_id, rank=rank)
        self.report(comm.NodeEvent(event_type=event_type, node=meta, reason=reason))

    def report_failure(self, error_data: str, level: str, restart_count: int = 0):
        self.report(
            comm.NodeFailure(
                node_id=self.node_id,
                error_data=error_data,
                level=level,
                restart_count=restart_count,
            )
        )

    def report_network_check_result(self, node_rank: int, normal: bool, elapsed: float):
        self.report(
            comm.NetworkCheckResult(
                node_id=node_rank, normal=normal, elapsed_time=elapsed
            )
        )

    def check_fault_node(self) -> Tuple[List[int], str]:
        resp = self.get(comm.NetworkCheckQuery(query=comm.NetworkCheckQuery.QUERY_FAULT))
        return resp.nodes, resp.reason

    def check_straggler(self) -> List[int]:
        resp = self.get(
            comm.NetworkCheckQuery(query=comm.NetworkCheckQuery.QUERY_STRAGGLER)
        )
        return resp.nodes

    def get_running_nodes(self) -> List[comm.NodeMeta]:
        resp = self.get(comm.RunningNodesRequest())
        return resp.nodes

    # -- data sharding -----------------------------------------------------------------

    def report_dataset_params(self, params: comm.DatasetShardParams):
        self.report(params)

    def get_task(self, dataset_name: str) -> comm.Task:
        resp = self.get(comm.TaskRequest(dataset_name=dataset_name, node_id=self.node_id))
        return resp if resp is not None else comm.Task()

    def report_task_result(self, dataset_name: str, task_id: int, success: bool = True,
                           err: str = ""):
        self.report(
            comm.TaskResult(
                dataset_name=dataset_name,
                task_id=task_id,
                node_id=self.node_id,
                success=success,
                err_message=err,
            )
        )

    def get_shard_checkpoint(self, dataset_name: str) -> str:
        resp = self.get(comm.ShardCheckpointRequest(dataset_name=dataset_name))
        return resp.content

    def report_shard_checkpoint(self, dataset_name: str, content: str):
        self.report(comm.ShardCheckpoint(dataset_name=dataset_name, content=content))

    # -- monitoring ----------------------------------------------------------------------

    def report_used_resource(self, stats: comm.ResourceStats):
        self.report(stats)

    def report_global_step(self, step: int, timestamp: float = 0.0):
        self.report(comm.GlobalStep(step=step, timestamp=timestamp or time.time()))

    def report_model_info(self, **kw):
        return self.report(comm.ModelInfo(**kw))

    def report_diagnosis_data(self, data_cls: str, content: str, node_rank: int = -1):
        self.report(
            comm.DiagnosisReportData(
                data_cls=data_cls,
                data_content=content,
                node_id=self.node_id,
                node_rank=node_rank,
            )
        )

    # -- config ---------------------------------------------------------------------------

    def get_paral_config(self) -> comm.ParallelConfig:
        return self.get(comm.ParallelConfigRequest())

    def get_elastic_run_config(self) -> Dict[str, str]:
        resp = self.get(comm.ElasticRunConfigRequest())
        return resp.configs

    def get_pre_check_result(self) -> comm.PreCheckResponse:
        return self.get(comm.PreCheckRequest(node_id=self.node_id))

    # -- sync ------------------------------------------------------------------------------

    def join_sync(self, sync_name: str):
        self.report(comm.SyncJoin(sync_name=sync_name, node_id=self.node_id))

    def sync_finished(self, sync_name: str):
        self.report(comm.SyncFinish(sync_name=sync_name))

    def is_sync_finished(self, sync_name: str) -> bool:
        resp = self.get(comm.SyncQuery(sync_name=sync_name))
        return resp.done

    def barrier(self, barrier_name: str, notify: bool = False) -> bool:
        resp = self.report(comm.BarrierRequest(barrier_name=barrier_name, notify=notify))
        return resp.done if resp else False

    def sync_checkpoint(self, step: int) -> bool:
        resp = self.get(comm.CkptSyncRequest(node_id=self.node_id, step=step))
        return resp.all_done
