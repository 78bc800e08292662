"""Rank-aware text logging (capability parity with
``scalerl/utils/logger/logging.py:30-106``: colored per-rank console logger,
file handler on rank 0 only, DDP de-duplication)."""

from __future__ import annotations

import logging
import os
import sys
from typing import Optional

_COLORS = {"DEBUG": "\033[36m", "INFO": "\033[32m", "WARNING": "\033[33m",
           "ERROR": "\033[31m", "CRITICAL": "\033[35m"}
_RESET = "\033[0m"


class _ColorFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        msg = super().format(record)
        if sys.stderr.isatty():
            color = _COLORS.get(record.levelname, "")
            return f"{color}{msg}{_RESET}"
        return msg


def get_rank() -> int:
    for var in ("RANK", "SLURM_PROCID"):
        if var in os.environ:
            return int(os.environ[var])
    return 0


def get_logger(name: str = "scalerl_amd",
               log_file: Optional[str] = None,
               level: int = logging.INFO,
               rank: Optional[int] = None) -> logging.Logger:
    """Per-rank logger: console everywhere (rank tag in the format),
    file handler attached on rank 0 only."""
    rank = get_rank() if rank is None else rank
    logger = logging.getLogger(name)
    if getattr(logger, "_scalerl_configured", False):
        return logger
    logger.setLevel(level if rank == 0 else max(level, logging.WARNING))
    fmt = f"%(asctime)s [rank{rank}] %(levelname)s %(name)s: %(message)s"
    ch = logging.StreamHandler(sys.stderr)
    ch.setFormatter(_ColorFormatter(fmt))
    logger.addHandler(ch)
    if log_file and rank == 0:
        os.makedirs(os.path.dirname(log_file) or ".", exist_ok=True)
        fh = logging.FileHandler(log_file)
        fh.setFormatter(logging.Formatter(fmt))
        logger.addHandler(fh)
    logger.propagate = False
    logger._scalerl_configured = True  # type: ignore[attr-defined]
    return logger
