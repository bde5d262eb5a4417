"""Structured logging: colored console + rotating file handler.

Reference parity: ``p2p/smart_node.py:28-124`` builds a colored console
logger with a custom VERBOSE level (5) and a ``TimedRotatingFileHandler``
writing ``logs/runtime.log`` daily with 7 backups (``debug_print``,
``smart_node.py:499-530`` adds role tags). Same surface here, minus the
socket-role plumbing: every component logs through ``get_logger`` and an
operator turns on file rotation with one call (tlctl does it for
``serve``).
"""

from __future__ import annotations

import logging
import logging.handlers
import os
from typing import Optional

VERBOSE = 5                      # below DEBUG, like the reference's level
logging.addLevelName(VERBOSE, "VERBOSE")

_COLORS = {
    logging.CRITICAL: "\x1b[91m",   # bright red
    logging.ERROR: "\x1b[31m",
    logging.WARNING: "\x1b[33m",
    logging.INFO: "\x1b[32m",
    logging.DEBUG: "\x1b[36m",
    VERBOSE: "\x1b[90m",
}
_RESET = "\x1b[0m"


class ColorFormatter(logging.Formatter):
    def __init__(self, use_color: Optional[bool] = None):
        super().__init__("%(asctime)s [%(tag)s] %(levelname)s %(message)s",
                         "%H:%M:%S")
        self.use_color = use_color

    def format(self, record):
        if not hasattr(record, "tag"):
            record.tag = record.name.rsplit(".", 1)[-1]
        msg = super().format(record)
        color_on = (self.use_color if self.use_color is not None
                    else os.isatty(2))
        if color_on:
            c = _COLORS.get(record.levelno, "")
            return f"{c}{msg}{_RESET}" if c else msg
        return msg


def get_logger(name: str = "tensorlink_amd",
               level: int = logging.INFO) -> logging.Logger:
    """Component logger with the colored console handler attached once.
    ``logger.verbose(...)`` logs at the custom VERBOSE level."""
    logger = logging.getLogger(name)
    if not getattr(logger, "_tl_configured", False):
        h = logging.StreamHandler()
        h.setFormatter(ColorFormatter())
        logger.addHandler(h)
        logger.setLevel(level)
        logger.propagate = False
        logger._tl_configured = True            # type: ignore[attr-defined]
        logger.verbose = (                       # type: ignore[attr-defined]
            lambda msg, *a, **kw: logger.log(VERBOSE, msg, *a, **kw))
    return logger


def enable_file_logging(log_dir: str = "logs",
                        filename: str = "runtime.log",
                        when: str = "midnight", backups: int = 7,
                        name: str = "tensorlink_amd") -> str:
    """Attach the reference's rotating file handler (daily x 7,
    ``smart_node.py:115-125``). Returns the log path."""
    os.makedirs(log_dir, exist_ok=True)
    path = os.path.join(log_dir, filename)
    logger = get_logger(name)
    for h in logger.handlers:
        if isinstance(h, logging.handlers.TimedRotatingFileHandler):
            return path
    fh = logging.handlers.TimedRotatingFileHandler(path, when=when,
                                                   backupCount=backups)
    fh.setFormatter(ColorFormatter(use_color=False))
    logger.addHandler(fh)
    return path
