"""ElasticJob-CRD scaler: instead of creating pods directly, patch a
ScalePlan custom resource and let the cluster operator reconcile it
(ref: dlrover/python/master/scaler/elasticjob_scaler.py; CRD types
go/elasticjob/api/v1alpha1).

The CRD API is injected like PodScaler's K8sApi so the sim harness/tests run
without a cluster. The ScalePlan body matches the reference operator's
schema so existing ElasticJob deployments reconcile it unchanged.
"""

from abc import ABC, abstractmethod
from typing import Dict, List, Optional

from dlrover_amd.common.log import logger
from dlrover_amd.common.node import Node

GROUP = "elastic.iml.github.io"
VERSION = "v1alpha1"
PLURAL = "scaleplans"


class CrdApi(ABC):
    @abstractmethod
    def apply_scaleplan(self, name: str, body: dict) -> bool:
        ...


class RealCrdApi(CrdApi):  # pragma: no cover - needs a cluster
    def __init__(self, namespace: str):
        from kubernetes import client, config

        config.load_incluster_config()
        self.namespace = namespace
        self.api = client.CustomObjectsApi()

    def apply_scaleplan(self, name: str, body: dict) -> bool:
        try:
            self.api.patch_namespaced_custom_object(
                GROUP, VERSION, self.namespace, PLURAL, name, body
            )
        except Exception:  # noqa: BLE001 - create if missing
            self.api.create_namespaced_custom_object(
                GROUP, VERSION, self.namespace, PLURAL, body
            )
        return True


class FakeCrdApi(CrdApi):
    def __init__(self):
        self.plans: List[dict] = []

    def apply_scaleplan(self, name: str, body: dict) -> bool:
        self.plans.append(body)
        return True


class ElasticJobScaler:
    """Emit ScalePlan CRs; the Go/argo operator creates/removes the pods."""

    def __init__(self, job_name: str, namespace: str = "default",
                 api: Optional[CrdApi] = None):
        self.job_name = job_name
        self.namespace = namespace
        self.api = api if api is not None else RealCrdApi(namespace)
        self._plan_index = 0

    def _plan_body(self, replicas: Dict[str, int],
                   remove_nodes: List[str]) -> dict:
        self._plan_index += 1
        return {
            "apiVersion": f"{GROUP}/{VERSION}",
            "kind": "ScalePlan",
            "metadata": {
                "name": f"{self.job_name}-scaleplan-{self._plan_index}",
                "namespace": self.namespace,
                "labels": {"elasticjob.dlrover/name": self.job_name},
            },
            "spec": {
                "ownerJob": self.job_name,
                "replicaResourceSpecs": {
                    role: {"replicas": count} for role, count in replicas.items()
                },
                "removePods": remove_nodes,
                "createdPods": [],
                "manualScaling": False,
            },
        }

    def launch_node(self, node: Node):
        body = self._plan_body({node.type: node.id + 1}, [])
        body["spec"]["createdPods"] = [
            {"name": f"{self.job_name}-{node.type}-{node.id}",
             "id": node.id, "type": node.type, "rankIndex": node.rank_index}
        ]
        self.api.apply_scaleplan(body["metadata"]["name"], body)
        logger.info("ScalePlan emitted: create %s-%s", node.type, node.id)

    def remove_node(self, node: Node):
        body = self._plan_body({}, [f"{self.job_name}-{node.type}-{node.id}"])
        self.api.apply_scaleplan(body["metadata"]["name"], body)
        node.is_released = True

    def scale_to(self, count: int, current_nodes: List[Node]):
        alive = [n for n in current_nodes if n.is_alive()]
        role = alive[0].type if alive else "worker"
        removes = []
        if len(alive) > count:
            removes = [
                f"{self.job_name}-{n.type}-{n.id}"
                for n in sorted(alive, key=lambda n: -n.rank_index)[: len(alive) - count]
            ]
        body = self._plan_body({role: count}, removes)
        self.api.apply_scaleplan(body["metadata"]["name"], body)
