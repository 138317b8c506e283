"""Rank-tagged logging (reference: worker.py:130-146 per-worker formatter)."""

import logging
import os
import sys

_FMT = "%(asctime)s %(levelname)s %(name)s: %(message)s"


def get_logger(name: str = "gllm_amd") -> logging.Logger:
    logger = logging.getLogger(name)
    if not logger.handlers:
        h = logging.StreamHandler(sys.stderr)
        rank = os.environ.get("RANK")
        tag = f"[rank{rank}] " if rank is not None else ""
        h.setFormatter(logging.Formatter(tag + _FMT, datefmt="%H:%M:%S"))
        logger.addHandler(h)
        logger.setLevel(os.environ.get("GLLM_LOG_LEVEL", "INFO"))
        logger.propagate = False
    return logger


logger = get_logger()
