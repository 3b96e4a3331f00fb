"""Logging setup (capability parity with reference
src/modalities/utils/logger_utils.py): a framework logger with rank-aware
formatting."""

import logging
import os

_LOGGER_NAME = "modalities_amd"


def get_logger(name: str = _LOGGER_NAME) -> logging.Logger:
    logger = logging.getLogger(name)
    if not logger.handlers:
        rank = os.environ.get("RANK", "0")
        handler = logging.StreamHandler()
        handler.setFormatter(logging.Formatter(
            f"%(asctime)s [rank {rank}] %(levelname)s %(name)s: %(message)s"))
        logger.addHandler(handler)
        logger.setLevel(os.environ.get("MODALITIES_AMD_LOG_LEVEL", "INFO"))
        logger.propagate = False
    return logger
