"""ElasticDataLoader: a DataLoader whose batch size the master can retune at
runtime (ref: dlrover/trainer/torch/elastic/dataloader.py + ParallelConfig
in comm.py:464-506)."""

from typing import Optional

from torch.utils.data import DataLoader

from dlrover_amd.common.log import logger


class ElasticDataLoader(DataLoader):
    def __init__(self, *args, config_version: int = 0, **kwargs):
        super().__init__(*args, **kwargs)
        self._config_version = config_version

    def update_batch_size(self, batch_size: Optional[int] = None):
        """Apply a master-pushed batch size (takes effect on next epoch's
        iterator). When batch_size is None, ask the master; VERSIONED
        suggestions are applied only once per version (--auto-tunning)."""
        if batch_size is None:
            try:
                from dlrover_amd.agent.master_client import MasterClient

                cfg = MasterClient.singleton_instance().get_paral_config()
                if cfg.dataloader.version <= self._config_version:
                    return
                self._config_version = cfg.dataloader.version
                batch_size = cfg.dataloader.batch_size or None
            except Exception:  # noqa: BLE001
                return
        if batch_size and batch_size > 0:
            # DataLoader freezes batch_size via the batch_sampler at init
            object.__setattr__(self, "batch_size", batch_size)
            if self.batch_sampler is not None:
                self.batch_sampler.batch_size = batch_size
            logger.info("dataloader batch size -> %s", batch_size)


    def maybe_autotune(self):
        """Poll the master's versioned suggestion when --auto-tunning is on
        (DLROVER_AUTO_TUNE); call at epoch boundaries."""
        import os

        if os.getenv("DLROVER_AUTO_TUNE", "") == "1":
            self.update_batch_size()
