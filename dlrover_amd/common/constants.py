"""Shared constants for the control plane.

Parity target: reference ``dlrover/python/common/constants.py`` (ref:
dlrover/python/common/constants.py:1-589) — node types/status/event names,
rendezvous names, accelerator types, default timeouts. Values that appear in
on-disk or on-wire formats (checkpoint tracker file name, done-file suffix)
are kept byte-identical so tooling written for the reference keeps working.
"""


class NodeType:
    MASTER = "master"
    WORKER = "worker"
    PS = "ps"
    EVALUATOR = "evaluator"
    CHIEF = "chief"


class NodeStatus:
    INITIAL = "Initial"
    PENDING = "Pending"
    RUNNING = "Running"
    SUCCEEDED = "Succeeded"
    FAILED = "Failed"
    DELETED = "Deleted"
    FINISHED = "Finished"
    BREAKDOWN = "Breakdown"
    UNKNOWN = "Unknown"

    @classmethod
    def terminal(cls):
        return {cls.SUCCEEDED, cls.FAILED, cls.DELETED, cls.FINISHED}


class NodeEventType:
    ADDED = "ADDED"
    MODIFIED = "MODIFIED"
    DELETED = "DELETED"
    # proactively reported by the agent
    NODE_CHECK_FAILED = "NODE_CHECK_FAILED"
    SUCCEEDED_EXITED = "SUCCEEDED_EXITED"
    FAILED_EXITED = "FAILED_EXITED"


class NodeExitReason:
    KILLED = "Killed"
    OOM = "OOMKilled"
    FATAL_ERROR = "Error"
    HARDWARE_ERROR = "HardwareError"
    RELAUNCHED = "Relaunched"
    NO_HEARTBEAT = "NoHeartBeat"
    UNKNOWN_ERROR = "UnknownError"


class JobStage:
    INIT = "INIT"
    RUNNING = "RUNNING"
    SUSPENDED = "SUSPENDED"
    STOPPING = "STOPPING"
    STOPPED = "STOPPED"


class JobExitReason:
    SUCCEEDED = "Succeeded"
    CODE_ERROR = "CodeError"
    WORKER_OOM = "WorkerOOM"
    WORKER_ERROR = "WorkerError"
    UNKNOWN_ERROR = "UnknownError"
    HANG_ERROR = "HangError"
    PENDING_TIMEOUT = "PendingTimeout"


class RendezvousName:
    """Two rendezvous planes, as in the reference (ref:
    dlrover/python/master/elastic_training/rdzv_manager.py)."""

    TRAINING = "elastic-training"
    NETWORK_CHECK = "network-check"


class Accelerators:
    """MI355X is the primary target; the generic names are kept for the
    launcher flag surface (ref: constants.py Accelerators)."""

    AMD_GPU = "amd.com/gpu"  # MI355X / gfx950 — first-class
    NVIDIA_GPU = "nvidia.com/gpu"
    GENERIC_CPU = "cpu"


class PlatformType:
    LOCAL = "local"
    KUBERNETES = "k8s"
    RAY = "ray"


class CommServiceType:
    TCP = "tcp"  # default: length-prefixed pickle over TCP (stdlib only)
    HTTP = "http"
    GRPC = "grpc"


class TrainingExceptionLevel:
    RPC_ERROR = "rpc_error"
    NODE_ERROR = "node_error"
    PROCESS_ERROR = "process_error"
    WARNING = "warning"
    INFO = "info"


class NetworkFailureReason:
    NO_INIT = "not_initialized"
    NODE_FAILURE = "node_failure"
    WAITING_NODE = "waiting_node"


class NodeEnv:
    """Environment variables understood by agent/worker processes."""

    MASTER_ADDR = "DLROVER_MASTER_ADDR"
    MASTER_SERVICE_TYPE = "DLROVER_MASTER_SERVICE_TYPE"
    NODE_ID = "NODE_ID"
    NODE_RANK = "NODE_RANK"
    NODE_NUM = "NODE_NUM"
    JOB_NAME = "ELASTIC_JOB_NAME"
    MONITOR_ENABLED = "DLROVER_MONITOR_ENABLED"
    # fault injection for tests (ref: node_check/utils.py:52 MOCK_ERR_RANK)
    MOCK_ERR_RANK = "MOCK_ERR_RANK"


class CheckpointConstant:
    """On-disk checkpoint layout constants — byte-compatible with the
    reference Flash Checkpoint layout (ref: ckpt_saver.py; storage tracker
    file 'dlrover_latest.txt')."""

    TRACKER_FILE = "dlrover_latest.txt"
    MODEL_STATES_NAME = "model_states"
    OPTIM_STATES_NAME = "optim_states"
    DONE_FILE_PREFIX = ".done_"
    SAVE_TIMEOUT = 600


class JobConstant:
    RENDEZVOUS_DEFAULT_INTERVAL = 1
    # seconds an agent may miss heartbeats before the master marks it dead
    NODE_HEARTBEAT_TIMEOUT = 120
    MASTER_CLIENT_TIMEOUT = 30
    MASTER_CLIENT_RETRY = 3
    TRAINING_AGENT_LOOP_INTERVAL = 5
    PRE_CHECK_WAIT_SECS = 5
    # hang detection: no step progress + hang metric for this long => hang
    HANG_DOWNTIME_SECS = 300


class GpuConstant:
    """MI355X (gfx950) node facts used by probes and thresholds.

    xGMI: each GPU has 7 point-to-point links at ~153 GB/s.
    A ring collective is per-link bound; RCCL engages multiple links. The
    node-check busbw thresholds below are calibrated for RCCL over xGMI,
    deliberately NOT for NVSwitch.
    """

    GPUS_PER_NODE = 8
    XGMI_LINKS_PER_GPU = 7
    XGMI_LINK_GBPS = 153.0
    HBM_GB = 288
    # bf16 matmul: guide-measured hipBLASLt ceiling ~2026 TF @8k; a healthy
    # node-check probe should clear a conservative fraction of that.
    BF16_MATMUL_HEALTHY_TFLOPS = 400.0
    # allreduce busbw floor for an 8-GPU xGMI node (single-process-per-GPU
    # RCCL ring ~ per-link bound; multi-channel lifts it).
    ALLREDUCE_HEALTHY_GBPS = 80.0


DEFAULT_MASTER_PORT = 24666
GRPC_MAX_MESSAGE_BYTES = 256 * 1024 * 1024
