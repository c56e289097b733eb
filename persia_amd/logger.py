"""Logging facade (mirrors reference persia/logger.py:1-128, without colorlog)."""
import logging
import os
import sys
from typing import Optional

_LOG_LEVEL = os.environ.get("LOG_LEVEL", "INFO").upper()
_DEFAULT_FMT = "%(asctime)s [%(levelname)s] %(name)s: %(message)s"

_loggers = {}


def get_logger(name: str, level: Optional[str] = None) -> logging.Logger:
    if name in _loggers:
        return _loggers[name]
    logger = logging.getLogger(name)
    logger.setLevel(level or _LOG_LEVEL)
    if not logger.handlers:
        handler = logging.StreamHandler(sys.stderr)
        handler.setFormatter(logging.Formatter(_DEFAULT_FMT))
        logger.addHandler(handler)
    logger.propagate = False
    _loggers[name] = logger
    return logger


def get_default_logger(name: str = "persia_amd") -> logging.Logger:
    return get_logger(name)
