"""Pod watcher: platform events -> NodeEvents for the job manager.

Parity target: ref dlrover/python/master/watcher/k8s_watcher.py:274-520
(PodWatcher: k8s watch stream -> NodeEvent; pod phase -> NodeStatus mapping).
The stream source is injected so tests drive it with a FakeEventSource.
"""

import queue
from typing import Iterator, Optional

from dlrover_amd.common import comm
from dlrover_amd.common.constants import NodeEventType, NodeStatus

_PHASE_TO_STATUS = {
    "Pending": NodeStatus.PENDING,
    "Running": NodeStatus.RUNNING,
    "Succeeded": NodeStatus.SUCCEEDED,
    "Failed": NodeStatus.FAILED,
    "Unknown": NodeStatus.UNKNOWN,
}


def pod_to_node_event(event_type: str, pod: dict) -> Optional[comm.NodeEvent]:
    """Translate one k8s pod event into our NodeEvent."""
    meta = pod.get("metadata", {})
    labels = meta.get("labels", {})
    try:
        node_id = int(labels.get("elasticjob.dlrover/replica-index", "-1"))
        rank = int(labels.get("elasticjob.dlrover/rank-index", node_id))
    except ValueError:
        return None
    if node_id < 0:
        return None
    phase = pod.get("status", {}).get("phase", "Unknown")
    status = _PHASE_TO_STATUS.get(phase, NodeStatus.UNKNOWN)
    if event_type == "DELETED":
        evt = NodeEventType.DELETED
    else:
        evt = event_type if event_type in ("ADDED", "MODIFIED") else "MODIFIED"
    # OOMKilled detection from container statuses (ref: k8s_watcher exit
    # reason parsing)
    reason = ""
    for cs in pod.get("status", {}).get("containerStatuses", []) or []:
        term = (cs.get("state") or {}).get("terminated") or {}
        if term.get("reason"):
            reason = term["reason"]
    node = comm.NodeMeta(
        type=labels.get("elasticjob.dlrover/replica-type", "worker"),
        id=node_id,
        rank=rank,
        status=status,
        addr=pod.get("status", {}).get("podIP", "") or "",
    )
    return comm.NodeEvent(event_type=evt, node=node, reason=reason)


class FakeEventSource:
    """Test double: push (event_type, pod) pairs; watcher consumes them."""

    def __init__(self):
        self._q: "queue.Queue" = queue.Queue()

    def push(self, event_type: str, pod: dict):
        self._q.put((event_type, pod))

    def stream(self) -> Iterator:
        while True:
            try:
                yield self._q.get(timeout=0.5)
            except queue.Empty:
                return


class K8sEventSource:  # pragma: no cover - needs a cluster
    def __init__(self, job_name: str, namespace: str):
        self.job_name = job_name
        self.namespace = namespace

    def stream(self) -> Iterator:
        from kubernetes import client, config, watch

        config.load_incluster_config()
        core = client.CoreV1Api()
        w = watch.Watch()
        selector = f"elasticjob.dlrover/name={self.job_name}"
        for event in w.stream(
            core.list_namespaced_pod,
            namespace=self.namespace,
            label_selector=selector,
            timeout_seconds=3600,
        ):
            yield event["type"], event["object"].to_dict()


class ScalePlanWatcher:
    """User-submitted ScalePlan CRs -> ResourcePlan for the job manager
    (ref: K8sScalePlanWatcher, k8s_watcher.py:354 — only ADDED manual-scale
    plans for THIS job are honored, each CR consumed once by uid).

    The event source is pluggable: FakeEventSource in the sim harness, a
    real custom-object watch stream on a cluster.
    """

    def __init__(self, job_name: str, namespace: str = "default", source=None):
        self.job_name = job_name
        self.namespace = namespace
        self._source = source or K8sCustomObjectSource(
            job_name, namespace, "scaleplans"
        )
        self._used_uids: set = set()

    def watch(self):
        from dlrover_amd.master.auto_scale import ResourcePlan

        for event_type, crd in self._source.stream():
            if event_type != "ADDED" or not crd:
                continue
            if crd.get("kind") != "ScalePlan":
                continue
            meta = crd.get("metadata", {})
            labels = meta.get("labels", {})
            if labels.get("elasticjob.dlrover/name", labels.get("job")) not in (
                None,
                self.job_name,
            ):
                continue
            if labels.get("scale-type", "manual") != "manual":
                continue
            uid = meta.get("uid") or meta.get("name")
            if uid in self._used_uids:
                continue
            self._used_uids.add(uid)
            spec = crd.get("spec", {})
            specs = spec.get("replicaResourceSpecs", {})
            worker = specs.get("worker", specs.get("Worker", {}))
            plan = ResourcePlan(
                node_count=int(worker.get("replicas", 0)),
                node_resource={
                    k: float(str(v).rstrip("mMiGg"))
                    for k, v in worker.get("resource", {}).items()
                },
                comment=f"scaleplan/{meta.get('name', uid)}",
            )
            yield plan


class K8sCustomObjectSource:  # pragma: no cover - needs a cluster
    """Watch stream over a custom-resource plural (ElasticJob / ScalePlan)."""

    def __init__(self, job_name: str, namespace: str, plural: str):
        self.job_name = job_name
        self.namespace = namespace
        self.plural = plural

    def stream(self) -> Iterator:
        from kubernetes import client, config, watch

        config.load_incluster_config()
        api = client.CustomObjectsApi()
        w = watch.Watch()
        for event in w.stream(
            api.list_namespaced_custom_object,
            "elastic.iml.github.io", "v1alpha1", self.namespace, self.plural,
            timeout_seconds=3600,
        ):
            yield event["type"], event["object"]


class ElasticJobWatcher:
    """Watches THIS job's ElasticJob CR for spec changes
    (ref: K8sElasticJobWatcher, k8s_watcher.py:450): a replica-count edit
    becomes a ResourcePlan, ``spec.suspend: true`` yields a stop signal.
    Yields ("scale", ResourcePlan) / ("suspend", None) / ("resume", None).
    """

    def __init__(self, job_name: str, namespace: str = "default", source=None):
        self.job_name = job_name
        self.namespace = namespace
        self._source = source or K8sCustomObjectSource(
            job_name, namespace, "elasticjobs"
        )
        self._last_replicas: Optional[int] = None
        self._suspended = False

    def watch(self):
        from dlrover_amd.master.auto_scale import ResourcePlan

        for event_type, crd in self._source.stream():
            if not crd or crd.get("kind") != "ElasticJob":
                continue
            if event_type not in ("ADDED", "MODIFIED"):
                continue
            meta = crd.get("metadata", {})
            if meta.get("name") not in (None, self.job_name):
                continue
            spec = crd.get("spec", {})
            suspended = bool(spec.get("suspend", False))
            if suspended != self._suspended:
                self._suspended = suspended
                yield ("suspend" if suspended else "resume", None)
            specs = spec.get("replicaSpecs", {})
            worker = specs.get("worker", specs.get("Worker", {}))
            replicas = worker.get("replicas")
            if replicas is None:
                continue
            replicas = int(replicas)
            if self._last_replicas is None:
                self._last_replicas = replicas  # initial spec, not a change
                continue
            if replicas != self._last_replicas:
                self._last_replicas = replicas
                yield (
                    "scale",
                    ResourcePlan(
                        node_count=replicas,
                        comment=f"elasticjob/{meta.get('name', self.job_name)}"
                                " replicas edit",
                    ),
                )


class PodWatcher:
    """Yields NodeEvents to DistributedJobManager._watch_events."""

    def __init__(self, job_name: str, namespace: str = "default", source=None):
        self.job_name = job_name
        self.source = source if source is not None else K8sEventSource(job_name, namespace)

    def watch(self) -> Iterator[comm.NodeEvent]:
        for event_type, pod in self.source.stream():
            evt = pod_to_node_event(event_type, pod)
            if evt is not None:
                yield evt
