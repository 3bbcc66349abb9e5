from dlrover_amd.trainer.flash_checkpoint.checkpointer import (  # noqa: F401
    Checkpointer,
    StorageType,
)
from dlrover_amd.trainer.flash_checkpoint.ddp import DdpCheckpointer  # noqa: F401
from dlrover_amd.trainer.flash_checkpoint.fsdp import FsdpShardCheckpointer  # noqa: F401
