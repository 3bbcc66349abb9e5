"""Logging setup for master/agent/worker roles (ref: dlrover/python/common/log.py)."""

import logging
import os
import sys

_FORMAT = "[%(asctime)s] [%(levelname)s] [%(name)s:%(lineno)d] %(message)s"


def _build_logger() -> logging.Logger:
    logger = logging.getLogger("dlrover_amd")
    if logger.handlers:
        return logger
    level = os.getenv("DLROVER_LOG_LEVEL", "INFO").upper()
    logger.setLevel(getattr(logging, level, logging.INFO))
    handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(logging.Formatter(_FORMAT))
    logger.addHandler(handler)
    log_dir = os.getenv("DLROVER_LOG_DIR", "")
    if log_dir:
        os.makedirs(log_dir, exist_ok=True)
        role = os.getenv("DLROVER_ROLE", "proc")
        fh = logging.FileHandler(os.path.join(log_dir, f"dlrover_{role}_{os.getpid()}.log"))
        fh.setFormatter(logging.Formatter(_FORMAT))
        logger.addHandler(fh)
    logger.propagate = False
    return logger


logger = _build_logger()
