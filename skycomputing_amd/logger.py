"""Per-rank logging.

Capability parity with the reference file logger (reference:
scaelum/logger/logger.py:4-14) on top of stdlib ``logging``: per-rank log
files plus console, flush-per-line semantics via a file handler.
"""

from __future__ import annotations

import logging
import os
import sys

_LOGGERS: dict[str, logging.Logger] = {}


class _FlushFileHandler(logging.FileHandler):
    def emit(self, record):
        super().emit(record)
        self.flush()


def get_logger(
    name: str = "skycomputing",
    log_file: str | None = None,
    level: int = logging.INFO,
    rank: int | None = None,
) -> logging.Logger:
    key = f"{name}:{log_file}"
    if key in _LOGGERS:
        return _LOGGERS[key]
    logger = logging.getLogger(key)
    logger.setLevel(level)
    logger.propagate = False
    fmt = logging.Formatter(
        f"%(levelname)s - %(asctime)s - rank{rank if rank is not None else '?'} - %(message)s"
    )
    sh = logging.StreamHandler(sys.stderr)
    sh.setFormatter(fmt)
    logger.addHandler(sh)
    if log_file is not None:
        os.makedirs(os.path.dirname(os.path.abspath(log_file)), exist_ok=True)
        fh = _FlushFileHandler(log_file, mode="a")
        fh.setFormatter(fmt)
        logger.addHandler(fh)
    _LOGGERS[key] = logger
    return logger


class Logger:
    """Minimal file logger with the reference's ``Logger.info`` surface."""

    def __init__(self, log_file: str | None = None, rank: int | None = None, name: str = "skycomputing"):
        self._logger = get_logger(name=name, log_file=log_file, rank=rank)

    def info(self, msg: str):
        self._logger.info(msg)

    def warning(self, msg: str):
        self._logger.warning(msg)

    def error(self, msg: str):
        self._logger.error(msg)
