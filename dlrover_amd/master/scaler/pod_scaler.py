"""Pod scaler: executes relaunch/scale decisions on Kubernetes.

Parity target: ref dlrover/python/master/scaler/pod_scaler.py:85-880 —
create/delete worker pods from a template, pending-create queue with a
background creator thread, service-per-pod. The k8s API is injected
(``K8sApi``) so the master runs against the real kubernetes client when
installed and against FakeK8sApi in tests / this container (the reference
tests use the same trick: tests/test_utils.py mocks k8sClient).
"""

import queue
import threading
import time
from abc import ABC, abstractmethod
from typing import Dict, List, Optional

from dlrover_amd.common.constants import NodeStatus, NodeType
from dlrover_amd.common.log import logger
from dlrover_amd.common.node import Node, NodeResource


class K8sApi(ABC):
    def create_service(self, svc_spec: dict) -> bool:  # pragma: no cover
        return True

    def delete_service(self, name: str) -> bool:  # pragma: no cover
        return True

    @abstractmethod
    def create_pod(self, pod_spec: dict) -> bool:
        ...

    @abstractmethod
    def delete_pod(self, name: str) -> bool:
        ...

    @abstractmethod
    def list_pods(self, label_selector: str) -> List[dict]:
        ...


class RealK8sApi(K8sApi):  # pragma: no cover - needs a cluster
    def __init__(self, namespace: str):
        from kubernetes import client, config

        config.load_incluster_config()
        self.namespace = namespace
        self.core = client.CoreV1Api()

    def create_pod(self, pod_spec: dict) -> bool:
        self.core.create_namespaced_pod(self.namespace, pod_spec)
        return True

    def delete_pod(self, name: str) -> bool:
        self.core.delete_namespaced_pod(name, self.namespace)
        return True

    def list_pods(self, label_selector: str) -> List[dict]:
        pods = self.core.list_namespaced_pod(
            self.namespace, label_selector=label_selector
        )
        return [p.to_dict() for p in pods.items]


class FakeK8sApi(K8sApi):
    """In-memory cluster for tests and the sim-master harness."""

    def __init__(self):
        self.pods: Dict[str, dict] = {}
        self.services: Dict[str, dict] = {}
        self.created: List[str] = []
        self.deleted: List[str] = []
        self._lock = threading.Lock()

    def create_pod(self, pod_spec: dict) -> bool:
        name = pod_spec["metadata"]["name"]
        with self._lock:
            self.pods[name] = pod_spec
            self.created.append(name)
        return True

    def delete_pod(self, name: str) -> bool:
        with self._lock:
            self.pods.pop(name, None)
            self.deleted.append(name)
        return True

    def create_service(self, svc_spec: dict) -> bool:
        with self._lock:
            self.services[svc_spec["metadata"]["name"]] = svc_spec
        return True

    def delete_service(self, name: str) -> bool:
        with self._lock:
            self.services.pop(name, None)
        return True

    def list_pods(self, label_selector: str) -> List[dict]:
        with self._lock:
            return list(self.pods.values())


def build_service_spec(job_name: str, node: Node) -> dict:
    """Stable per-pod Service (ref: pod_scaler.py:776 service-per-pod):
    gives every worker a DNS name that survives pod relaunch, so anything
    addressing workers by name (sidecars, debuggers, the dashboard) does
    not chase pod IPs."""
    name = f"{job_name}-{node.type}-{node.rank_index}"
    return {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": {
            "name": name,
            "labels": {"elasticjob.dlrover/name": job_name},
        },
        "spec": {
            "clusterIP": "None",  # headless
            "selector": {
                "elasticjob.dlrover/name": job_name,
                "elasticjob.dlrover/replica-type": node.type,
                "elasticjob.dlrover/rank-index": str(node.rank_index),
            },
            "ports": [{"name": "master", "port": 22222}],
        },
    }


def build_pod_spec(
    job_name: str,
    node: Node,
    image: str = "dlrover-amd:latest",
    command: Optional[List[str]] = None,
    master_addr: str = "",
) -> dict:
    """Worker pod template (ref: pod_scaler.py:190 _create_pod_template):
    MI355X resource key amd.com/gpu; env carries node identity + master."""
    res = node.config_resource
    gpu = {"amd.com/gpu": str(res.gpu_num)} if res.gpu_num else {}
    return {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": {
            "name": f"{job_name}-{node.type}-{node.id}",
            "labels": {
                "elasticjob.dlrover/name": job_name,
                "elasticjob.dlrover/replica-type": node.type,
                "elasticjob.dlrover/replica-index": str(node.id),
                "elasticjob.dlrover/rank-index": str(node.rank_index),
            },
        },
        "spec": {
            "restartPolicy": "Never",
            "containers": [
                {
                    "name": "main",
                    "image": image,
                    "command": command or ["/bin/bash", "-c", "dlrover-run"],
                    "env": [
                        {"name": "NODE_ID", "value": str(node.id)},
                        {"name": "NODE_RANK", "value": str(node.rank_index)},
                        {"name": "DLROVER_MASTER_ADDR", "value": master_addr},
                        {"name": "ELASTIC_JOB_NAME", "value": job_name},
                        # dmabuf IPC is required for RCCL cross-process
                        {"name": "HSA_ENABLE_IPC_MODE_LEGACY", "value": "0"},
                    ],
                    "resources": {
                        "limits": {
                            "cpu": str(res.cpu or 8),
                            "memory": f"{res.memory_mb or 65536}Mi",
                            **gpu,
                        }
                    },
                }
            ],
        },
    }


class PodScaler:
    """Create/remove worker pods; a background thread drains the create queue
    (ref: pod_scaler.py:533-579)."""

    def __init__(
        self,
        job_name: str,
        namespace: str = "default",
        api: Optional[K8sApi] = None,
        master_addr: str = "",
        image: str = "dlrover-amd:latest",
        service_per_pod: bool = False,
    ):
        self.job_name = job_name
        self.namespace = namespace
        self.api = api if api is not None else RealK8sApi(namespace)
        self.master_addr = master_addr
        self.image = image
        self.service_per_pod = service_per_pod
        self._create_q: "queue.Queue[Node]" = queue.Queue()
        self._stop = threading.Event()
        self._thread = threading.Thread(
            target=self._creator_loop, name="pod-creator", daemon=True
        )
        self._thread.start()

    def launch_node(self, node: Node):
        self._create_q.put(node)

    def remove_node(self, node: Node):
        name = f"{self.job_name}-{node.type}-{node.id}"
        try:
            self.api.delete_pod(name)
            node.is_released = True
        except Exception:  # noqa: BLE001
            logger.exception("delete pod %s failed", name)

    def scale_to(self, count: int, current_nodes: List[Node]):
        """Scale the worker group to `count` (ref: ScalePlan execution)."""
        alive = [n for n in current_nodes if n.is_alive()]
        if len(alive) < count:
            # pod ids only ever grow, but rank indices must re-fill the holes
            # a previous scale-down left so the final rank space is a
            # contiguous 0..count-1 (torchrun node ranks)
            base = max((n.id for n in current_nodes), default=-1) + 1
            used_ranks = {n.rank_index for n in alive}
            free_ranks = [r for r in range(count) if r not in used_ranks]
            for i, rank in enumerate(free_ranks[: count - len(alive)]):
                node = Node(
                    NodeType.WORKER,
                    base + i,
                    rank_index=rank,
                    config_resource=NodeResource(gpu_num=8),
                )
                self.launch_node(node)
        elif len(alive) > count:
            # drop the highest ranks first (keeps rank contiguity)
            for node in sorted(alive, key=lambda n: -n.rank_index)[: len(alive) - count]:
                self.remove_node(node)

    def _creator_loop(self):
        while not self._stop.is_set():
            try:
                node = self._create_q.get(timeout=1.0)
            except queue.Empty:
                continue
            spec = build_pod_spec(
                self.job_name, node, image=self.image, master_addr=self.master_addr
            )
            for attempt in range(3):
                try:
                    self.api.create_pod(spec)
                    if self.service_per_pod:
                        self.api.create_service(
                            build_service_spec(self.job_name, node)
                        )
                    node.update_status(NodeStatus.PENDING)
                    break
                except Exception:  # noqa: BLE001
                    logger.exception(
                        "create pod attempt %s failed for node %s", attempt, node.id
                    )
                    time.sleep(2 ** attempt)

    def stop(self):
        self._stop.set()
        self._thread.join(timeout=3)
