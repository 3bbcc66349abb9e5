"""Platform job-args: resolve a JobArgs description from the environment /
ElasticJob CR / defaults (ref: dlrover/python/scheduler/{job,kubernetes,
factory}.py — JobArgs, new_job_args)."""

import os
from dataclasses import dataclass, field
from typing import Dict, Optional

from dlrover_amd.common.constants import PlatformType
from dlrover_amd.common.node import NodeGroupResource, NodeResource


@dataclass
class JobArgs:
    platform: str = PlatformType.LOCAL
    job_name: str = "dlrover-job"
    namespace: str = "default"
    node_groups: Dict[str, NodeGroupResource] = field(default_factory=dict)
    distribution_strategy: str = "AllreduceStrategy"
    relaunch_on_worker_failure: int = 3
    remove_exited_node: bool = True
    cordon_fault_node: bool = True

    @property
    def worker_count(self) -> int:
        g = self.node_groups.get("worker")
        return g.count if g else 1


def job_args_from_env() -> JobArgs:
    args = JobArgs(
        platform=os.getenv("DLROVER_PLATFORM", PlatformType.LOCAL),
        job_name=os.getenv("ELASTIC_JOB_NAME", "dlrover-job"),
        namespace=os.getenv("DLROVER_NAMESPACE", "default"),
    )
    args.node_groups["worker"] = NodeGroupResource(
        count=int(os.getenv("NODE_NUM", "1")),
        node_resource=NodeResource(
            cpu=float(os.getenv("DLROVER_WORKER_CPU", "8")),
            memory_mb=int(os.getenv("DLROVER_WORKER_MEM_MB", "65536")),
            gpu_num=int(os.getenv("DLROVER_WORKER_GPU", "8")),
            gpu_type="amd.com/gpu",
        ),
    )
    return args


def job_args_from_elasticjob_cr(cr: dict) -> JobArgs:
    """Parse an ElasticJob custom resource dict (the operator's spec shape:
    go/elasticjob/api/v1alpha1/elasticjob_types.go:26-124)."""
    meta = cr.get("metadata", {})
    spec = cr.get("spec", {})
    args = JobArgs(
        platform=PlatformType.KUBERNETES,
        job_name=meta.get("name", "dlrover-job"),
        namespace=meta.get("namespace", "default"),
        distribution_strategy=spec.get("distributionStrategy", "AllreduceStrategy"),
    )
    for role, rs in (spec.get("replicaSpecs") or {}).items():
        replicas = int(rs.get("replicas", 1))
        res = {}
        try:
            containers = rs["template"]["spec"]["containers"]
            res = containers[0].get("resources", {}).get("limits", {}) or {}
        except (KeyError, IndexError, TypeError):
            pass
        args.node_groups[role] = NodeGroupResource(
            count=replicas,
            node_resource=NodeResource(
                cpu=_parse_cpu(res.get("cpu", 8)),
                memory_mb=_parse_mem(res.get("memory", "65536Mi")),
                gpu_num=int(res.get("amd.com/gpu", res.get("nvidia.com/gpu", 0)) or 0),
                gpu_type="amd.com/gpu" if "amd.com/gpu" in res else "",
            ),
        )
    return args


def _parse_mem(v) -> int:
    s = str(v)
    try:
        if s.endswith("Gi"):
            return int(float(s[:-2]) * 1024)
        if s.endswith("Mi"):
            return int(float(s[:-2]))
        if s.endswith("Ki"):
            return int(float(s[:-2]) / 1024)
        return int(float(s) / (1 << 20))
    except ValueError:
        return 65536


def _parse_cpu(v) -> float:
    """k8s CPU quantity: '500m' = 0.5 cores, '4' = 4 cores."""
    s = str(v)
    try:
        if s.endswith("m"):
            return float(s[:-1]) / 1000.0
        return float(s)
    except ValueError:
        return 8.0


def new_job_args(platform: str, cr: Optional[dict] = None) -> JobArgs:
    if platform == PlatformType.KUBERNETES and cr is not None:
        return job_args_from_elasticjob_cr(cr)
    return job_args_from_env()
