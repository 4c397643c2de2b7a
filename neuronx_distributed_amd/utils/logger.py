"""Rank-aware logger (reference: utils/logger.py:17-52 — NXD_LOG_LEVEL env
control + rank-0-only default filter)."""

import logging
import os
import sys

_LOGGERS = {}


def _level_from_env() -> int:
    lvl = os.environ.get("NXDA_LOG_LEVEL", os.environ.get("NXD_LOG_LEVEL", "INFO"))
    return getattr(logging, lvl.upper(), logging.INFO)


class _Rank0Filter(logging.Filter):
    def filter(self, record):
        if os.environ.get("NXDA_LOG_ALL_RANKS", "0") == "1":
            return True
        rank = os.environ.get("RANK")
        return rank is None or rank == "0"


def get_logger(name: str = "nxd_amd", rank0_only: bool = True) -> logging.Logger:
    key = (name, rank0_only)
    if key in _LOGGERS:
        return _LOGGERS[key]
    logger = logging.getLogger(name)
    logger.setLevel(_level_from_env())
    logger.propagate = False
    if not logger.handlers:
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(
            logging.Formatter("[%(asctime)s %(levelname)s %(name)s] %(message)s")
        )
        logger.addHandler(h)
    if rank0_only:
        logger.addFilter(_Rank0Filter())
    _LOGGERS[key] = logger
    return logger
