"""Logging helpers (reference uses tf_logging throughout)."""

import logging
import os
import sys

_LOGGER = None


def get_logger():
    global _LOGGER
    if _LOGGER is None:
        logger = logging.getLogger("epl_amd")
        if not logger.handlers:
            handler = logging.StreamHandler(sys.stderr)
            rank = os.environ.get("RANK", "0")
            handler.setFormatter(logging.Formatter(
                "[epl-amd r{}] %(levelname)s %(message)s".format(rank)))
            logger.addHandler(handler)
        logger.setLevel(os.environ.get("EPL_LOG_LEVEL", "INFO"))
        _LOGGER = logger
    return _LOGGER
