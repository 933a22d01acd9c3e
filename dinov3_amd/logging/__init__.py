"""Logging setup: glog-style format, rank-0 stdout + per-rank files.

Capability parity with the reference (dinov3_jax/logging/__init__.py:29-150)
without the termcolor dependency.
"""

from __future__ import annotations

import functools
import logging
import os
import sys
from typing import Optional

from .helpers import MetricLogger, SmoothedValue  # noqa: F401


class _GlogFormatter(logging.Formatter):
    LEVEL_CHAR = {
        logging.DEBUG: "D",
        logging.INFO: "I",
        logging.WARNING: "W",
        logging.ERROR: "E",
        logging.CRITICAL: "C",
    }

    def format(self, record: logging.LogRecord) -> str:
        level = self.LEVEL_CHAR.get(record.levelno, "?")
        date = self.formatTime(record, "%Y%m%d %H:%M:%S")
        prefix = f"{level}{date} {record.process} {record.name} {record.filename}:{record.lineno}]"
        return f"{prefix} {record.getMessage()}" + (
            "\n" + self.formatException(record.exc_info) if record.exc_info else ""
        )


@functools.lru_cache(maxsize=1)
def _rank() -> int:
    from .. import parallel

    return parallel.get_rank()


def setup_logging(output: Optional[str] = None, name: str = "dinov3",
                  level: int = logging.INFO) -> logging.Logger:
    logger = logging.getLogger(name)
    logger.setLevel(level)
    logger.propagate = False
    if logger.handlers:
        return logger
    fmt = _GlogFormatter()
    if _rank() == 0:
        sh = logging.StreamHandler(stream=sys.stdout)
        sh.setFormatter(fmt)
        logger.addHandler(sh)
    if output:
        os.makedirs(output, exist_ok=True)
        path = os.path.join(output, f"log.rank{_rank()}.txt" if _rank() else "log.txt")
        fh = logging.FileHandler(path)
        fh.setFormatter(fmt)
        logger.addHandler(fh)
    if not logger.handlers:  # non-main rank, no output dir: swallow
        logger.addHandler(logging.NullHandler())
    return logger
