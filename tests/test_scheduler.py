"""JobArgs resolution from env and ElasticJob CRs."""

from dlrover_amd.scheduler import job_args_from_elasticjob_cr, new_job_args


def test_job_args_from_env(monkeypatch):
    monkeypatch.setenv("NODE_NUM", "4")
    monkeypatch.setenv("DLROVER_WORKER_GPU", "8")
    args = new_job_args("local")
    assert args.worker_count == 4
    assert args.node_groups["worker"].node_resource.gpu_num == 8


def test_job_args_from_cr():
    cr = {
        "metadata": {"name": "llama-job", "namespace": "train"},
        "spec": {
            "distributionStrategy": "AllreduceStrategy",
            "replicaSpecs": {
                "worker": {
                    "replicas": 16,
                    "template": {"spec": {"containers": [{
                        "resources": {"limits": {
                            "cpu": "96", "memory": "1024Gi", "amd.com/gpu": "8",
                        }}
                    }]}},
                }
            },
        },
    }
    args = job_args_from_elasticjob_cr(cr)
    assert args.job_name == "llama-job" and args.namespace == "train"
    g = args.node_groups["worker"]
    assert g.count == 16
    assert g.node_resource.gpu_num == 8
    assert g.node_resource.memory_mb == 1024 * 1024
    assert g.node_resource.gpu_type == "amd.com/gpu"


def test_cpu_quantity_parsing():
    """k8s millicore CPU quantities: '500m' is half a core, not 500."""
    from dlrover_amd.scheduler import job_args_from_elasticjob_cr

    cr = {
        "metadata": {"name": "j"},
        "spec": {"replicaSpecs": {"worker": {
            "replicas": 2,
            "template": {"spec": {"containers": [{
                "resources": {"limits": {"cpu": "500m", "memory": "2Gi",
                                         "amd.com/gpu": 1}}
            }]}},
        }}},
    }
    g = job_args_from_elasticjob_cr(cr).node_groups["worker"]
    assert g.node_resource.cpu == 0.5
    assert g.node_resource.memory_mb == 2048
    assert g.node_resource.gpu_num == 1
