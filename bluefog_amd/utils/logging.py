# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Leveled logging (reference analog: bluefog/common/logging.{h,cc} and the
python "bluefog" logger, basics.py:27-34). Level comes from
``BLUEFOG_LOG_LEVEL`` (trace/debug/info/warn/error/fatal);
``BLUEFOG_LOG_HIDE_TIME=1`` drops timestamps."""

import logging
import os
import sys

_LEVELS = {
    "trace": logging.DEBUG,
    "debug": logging.DEBUG,
    "info": logging.INFO,
    "warn": logging.WARNING,
    "warning": logging.WARNING,
    "error": logging.ERROR,
    "fatal": logging.CRITICAL,
}

_logger = None


def get_logger() -> logging.Logger:
    global _logger
    if _logger is not None:
        return _logger
    logger = logging.getLogger("bluefog_amd")
    level = _LEVELS.get(os.environ.get("BLUEFOG_LOG_LEVEL", "warn").lower(), logging.WARNING)
    logger.setLevel(level)
    if not logger.handlers:
        handler = logging.StreamHandler(sys.stderr)
        rank = os.environ.get("RANK", "0")
        if os.environ.get("BLUEFOG_LOG_HIDE_TIME", "0") == "1":
            fmt = f"[bf rank {rank}] %(levelname)s %(message)s"
        else:
            fmt = f"%(asctime)s [bf rank {rank}] %(levelname)s %(message)s"
        handler.setFormatter(logging.Formatter(fmt))
        logger.addHandler(handler)
    logger.propagate = False
    _logger = logger
    return logger
