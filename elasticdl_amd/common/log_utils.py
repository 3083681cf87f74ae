"""Logger factory (reference: elasticdl/python/common/log_utils.py)."""

import logging
import os
import sys

_FORMAT = "[%(asctime)s] [%(levelname)s] [%(name)s:%(lineno)d] %(message)s"


def get_logger(name: str, level: str = None) -> logging.Logger:
    logger = logging.getLogger(name)
    if not logger.handlers:
        handler = logging.StreamHandler(sys.stderr)
        handler.setFormatter(logging.Formatter(_FORMAT))
        logger.addHandler(handler)
        logger.propagate = False
    logger.setLevel(level or os.environ.get("EDL_LOG_LEVEL", "INFO"))
    return logger


default_logger = get_logger("elasticdl_amd")
