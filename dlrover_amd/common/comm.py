"""Control-plane RPC message types.

The agent↔master protocol is two verbs — ``report(msg)`` (fire-and-ack) and
``get(msg)`` (request/response) — carrying typed dataclasses, exactly the
shape of the reference protocol (ref: dlrover/proto/elastic_training.proto:29-33,
dlrover/python/common/comm.py:105-560). Transport is pluggable (tcp/http/grpc,
see dlrover_amd.utils.transport); messages are pickled with the restricted
unpickler in dlrover_amd.common.serialize.
"""

import socket
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class Message:
    """Base class — every RPC payload derives from this."""


# ---------------------------------------------------------------------------
# envelope
# ---------------------------------------------------------------------------


@dataclass
class BaseRequest:
    node_id: int = -1
    node_type: str = ""
    data: Optional[Message] = None


@dataclass
class BaseResponse:
    success: bool = True
    reason: str = ""
    data: Optional[Message] = None


# ---------------------------------------------------------------------------
# rendezvous
# ---------------------------------------------------------------------------


@dataclass
class JoinRendezvousRequest(Message):
    node_id: int = 0
    node_rank: int = 0
    local_world_size: int = 1
    rdzv_name: str = ""
    node_ip: str = ""


@dataclass
class JoinRendezvousResponse(Message):
    round: int = 0


@dataclass
class CommWorldRequest(Message):
    node_id: int = 0
    rdzv_name: str = ""


@dataclass
class CommWorldResponse(Message):
    rdzv_round: int = 0
    group: int = 0
    # node_rank -> local_world_size, sorted by topology
    world: Dict[int, int] = field(default_factory=dict)


@dataclass
class RdzvBlockRequest(Message):
    """Take/release a completion hold on the pending rendezvous round while
    this node persists shards for a UCP reshard (ref: UcpRdzvManager)."""

    node_rank: int = 0
    blocked: bool = True
    rdzv_name: str = ""


@dataclass
class WaitingNodeNumRequest(Message):
    node_id: int = 0
    local_world_size: int = 1
    rdzv_name: str = ""


@dataclass
class WaitingNodeNumResponse(Message):
    waiting_num: int = 0


@dataclass
class RendezvousParams(Message):
    min_nodes: int = 1
    max_nodes: int = 1
    waiting_timeout: float = 60.0
    node_unit: int = 1
    joint_timeout: float = 600.0


# ---------------------------------------------------------------------------
# KV store (backs MasterKVStore = torch Store for process-group bootstrap)
# ---------------------------------------------------------------------------


@dataclass
class KeyValuePair(Message):
    key: str = ""
    value: bytes = b""


@dataclass
class KeyValuePairs(Message):
    kvs: Dict[str, bytes] = field(default_factory=dict)


@dataclass
class KVStoreGetRequest(Message):
    key: str = ""


@dataclass
class KVStoreMultiGetRequest(Message):
    keys: List[str] = field(default_factory=list)


@dataclass
class KVStoreAddRequest(Message):
    key: str = ""
    amount: int = 0


@dataclass
class KVStoreAddResponse(Message):
    value: int = 0


@dataclass
class KVStoreDeleteRequest(Message):
    key: str = ""


# ---------------------------------------------------------------------------
# node lifecycle / health
# ---------------------------------------------------------------------------


@dataclass
class NodeMeta(Message):
    type: str = ""
    id: int = 0
    rank: int = 0
    addr: str = ""
    status: str = ""
    cpu: float = 0.0
    memory_mb: int = 0
    gpu_num: int = 0
    gpu_type: str = ""


@dataclass
class NodeEvent(Message):
    event_type: str = ""
    node: Optional[NodeMeta] = None
    reason: str = ""


@dataclass
class HeartbeatRequest(Message):
    node_id: int = 0
    node_rank: int = -1
    timestamp: float = 0.0


@dataclass
class HeartbeatResponse(Message):
    # a master-pushed DiagnosisAction serialized as (cls_name, kwargs)
    action_cls: str = ""
    action_kwargs: dict = field(default_factory=dict)


@dataclass
class NodeFailure(Message):
    node_id: int = 0
    node_rank: int = -1
    error_data: str = ""
    level: str = ""
    restart_count: int = 0


@dataclass
class ResourceStats(Message):
    node_id: int = 0
    cpu_percent: float = 0.0
    memory_mb: int = 0
    gpu_stats: List[dict] = field(default_factory=list)


@dataclass
class GlobalStep(Message):
    step: int = 0
    timestamp: float = 0.0


@dataclass
class NetworkCheckResult(Message):
    node_id: int = 0
    normal: bool = True
    elapsed_time: float = 0.0
    round: int = 0


@dataclass
class NetworkCheckQuery(Message):
    QUERY_FAULT = "fault"
    QUERY_STRAGGLER = "straggler"
    query: str = "fault"


@dataclass
class NetworkCheckReply(Message):
    nodes: List[int] = field(default_factory=list)
    reason: str = ""


@dataclass
class RunningNodesRequest(Message):
    pass


@dataclass
class RunningNodes(Message):
    nodes: List[NodeMeta] = field(default_factory=list)


# ---------------------------------------------------------------------------
# data sharding
# ---------------------------------------------------------------------------


@dataclass
class DatasetShardParams(Message):
    dataset_name: str = ""
    dataset_size: int = 0
    shard_size: int = 0
    batch_size: int = 0
    num_epochs: int = 1
    shuffle: bool = False
    storage_type: str = ""
    splitter: str = "batch"


@dataclass
class TaskRequest(Message):
    dataset_name: str = ""
    node_id: int = 0


@dataclass
class Task(Message):
    task_id: int = -1
    task_type: str = ""
    dataset_name: str = ""
    shard_name: str = ""
    start: int = 0
    end: int = 0
    epoch: int = 0

    @property
    def empty(self) -> bool:
        return self.task_id < 0


@dataclass
class TaskResult(Message):
    dataset_name: str = ""
    task_id: int = -1
    node_id: int = 0
    success: bool = True
    err_message: str = ""


@dataclass
class ShardCheckpointRequest(Message):
    dataset_name: str = ""


@dataclass
class ShardCheckpoint(Message):
    dataset_name: str = ""
    content: str = ""


# ---------------------------------------------------------------------------
# elastic run config / parallelism tuning
# ---------------------------------------------------------------------------


@dataclass
class DataLoaderConfig(Message):
    dataloader_name: str = ""
    batch_size: int = 0
    num_workers: int = 0
    pin_memory: bool = True
    version: int = 0  # trainers apply only configs newer than theirs


@dataclass
class OptimizerConfig(Message):
    optimizer_name: str = ""
    learning_rate: float = 0.0
    version: int = 0


@dataclass
class ModelInfo(Message):
    """Trainer-reported model card (ref: stats/training_metrics model info):
    feeds the strategy generator's activation-memory estimate and the
    dashboard."""

    model_name: str = ""
    params: int = 0
    n_layers: int = 0
    n_heads: int = 0
    hidden_size: int = 0
    seq_len: int = 0
    dtype: str = ""


@dataclass
class ParallelConfigRequest(Message):
    pass


@dataclass
class ParallelConfig(Message):
    dataloader: DataLoaderConfig = field(default_factory=DataLoaderConfig)
    optimizer: OptimizerConfig = field(default_factory=OptimizerConfig)
    restart: bool = False


@dataclass
class ElasticRunConfigRequest(Message):
    pass


@dataclass
class ElasticRunConfig(Message):
    configs: Dict[str, str] = field(default_factory=dict)


@dataclass
class PreCheckRequest(Message):
    node_id: int = 0


@dataclass
class PreCheckResponse(Message):
    PASS = "PASS"
    FAIL = "FAIL"
    CHECKING = "CHECKING"
    status: str = PASS
    reason: str = ""


# ---------------------------------------------------------------------------
# sync / barrier
# ---------------------------------------------------------------------------


@dataclass
class SyncJoin(Message):
    sync_name: str = ""
    node_id: int = 0
    node_rank: int = 0


@dataclass
class SyncFinish(Message):
    sync_name: str = ""


@dataclass
class SyncQuery(Message):
    sync_name: str = ""


@dataclass
class SyncResult(Message):
    done: bool = False


@dataclass
class BarrierRequest(Message):
    barrier_name: str = ""
    notify: bool = False


# ---------------------------------------------------------------------------
# diagnosis
# ---------------------------------------------------------------------------


@dataclass
class DiagnosisReportData(Message):
    data_cls: str = ""
    data_content: str = ""
    node_id: int = 0
    node_type: str = ""
    node_rank: int = -1


@dataclass
class TrainingStatusRequest(Message):
    pass


@dataclass
class TrainingStatusReply(Message):
    status: str = ""


# ---------------------------------------------------------------------------
# checkpoint coordination
# ---------------------------------------------------------------------------


@dataclass
class CkptSyncRequest(Message):
    """Master-side consensus on which step every shard finished persisting
    (ref: rdzv_manager.sync_ckpt_nodes)."""

    node_id: int = 0
    step: int = 0


@dataclass
class CkptSyncResponse(Message):
    all_done: bool = False


def addr_connectable(addr: str, timeout: float = 1.0) -> bool:
    """True if host:port accepts a TCP connection."""
    try:
        host, port = addr.rsplit(":", 1)
        with socket.create_connection((host, int(port)), timeout=timeout):
            return True
    except (OSError, ValueError):
        return False


def now() -> float:
    return time.time()
