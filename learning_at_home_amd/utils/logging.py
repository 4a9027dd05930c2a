"""Per-module loggers (reference observability = `logging` only)."""
import logging
import os

_FMT = "[%(asctime)s.%(msecs)03d][%(levelname)s][%(name)s] %(message)s"
_DATEFMT = "%H:%M:%S"
_configured = False


def get_logger(name: str) -> logging.Logger:
    global _configured
    if not _configured:
        level = os.environ.get("LAH_LOG_LEVEL", "INFO").upper()
        logging.basicConfig(format=_FMT, datefmt=_DATEFMT, level=getattr(logging, level, logging.INFO))
        _configured = True
    return logging.getLogger(name)
