"""Structured logging (reference parity: pkg/logutil/logutil.go:10-33 —
zap dev config, colored level, `app` field, Info unless verbose)."""

from __future__ import annotations

import logging
import sys

_COLORS = {
    logging.DEBUG: "\x1b[36m", logging.INFO: "\x1b[32m",
    logging.WARNING: "\x1b[33m", logging.ERROR: "\x1b[31m",
    logging.CRITICAL: "\x1b[35m",
}
_RESET = "\x1b[0m"


class _Fmt(logging.Formatter):
    def __init__(self, color: bool):
        super().__init__()
        self.color = color

    def format(self, record: logging.LogRecord) -> str:
        lvl = record.levelname
        if self.color:
            lvl = f"{_COLORS.get(record.levelno, '')}{lvl}{_RESET}"
        base = (f"{self.formatTime(record, '%H:%M:%S.%f'[:-3])} {lvl} "
                f"[{record.name}] {record.getMessage()}")
        if record.exc_info:
            base += "\n" + self.formatException(record.exc_info)
        return base


def new_app_logger(app: str, verbose: bool = False) -> logging.Logger:
    logger = logging.getLogger(app)
    if not logger.handlers:
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(_Fmt(color=sys.stderr.isatty()))
        logger.addHandler(h)
        logger.propagate = False
    logger.setLevel(logging.DEBUG if verbose else logging.INFO)
    return logger
