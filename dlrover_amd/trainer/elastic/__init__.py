from dlrover_amd.trainer.elastic.sampler import ElasticDistributedSampler  # noqa: F401
from dlrover_amd.trainer.elastic.trainer import ElasticTrainer  # noqa: F401
from dlrover_amd.trainer.elastic.dataloader import ElasticDataLoader  # noqa: F401
