"""Structured JSON logger (zap-equivalent; LOG_LEVEL env honored).

Ref internal/logger/logger.go:14-54.
"""
from __future__ import annotations

import json
import logging
import os
import sys
import time
from typing import Any, Optional

_LEVELS = {
    "debug": logging.DEBUG,
    "info": logging.INFO,
    "warn": logging.WARNING,
    "error": logging.ERROR,
}


class JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        entry: dict[str, Any] = {
            "level": record.levelname.lower(),
            "ts": time.time(),
            "caller": f"{record.module}:{record.lineno}",
            "msg": record.getMessage(),
        }
        extra = getattr(record, "kv", None)
        if extra:
            entry.update(extra)
        if record.exc_info and record.exc_info[0] is not None:
            entry["error"] = self.formatException(record.exc_info)
        return json.dumps(entry)


_logger: Optional[logging.Logger] = None


def init_logger(name: str = "inferno-amd") -> logging.Logger:
    global _logger
    if _logger is not None:
        return _logger
    level = _LEVELS.get(os.environ.get("LOG_LEVEL", "info").lower(), logging.INFO)
    logger = logging.getLogger(name)
    logger.setLevel(level)
    handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(JsonFormatter())
    logger.addHandler(handler)
    logger.propagate = False
    _logger = logger
    return logger


def log() -> logging.Logger:
    return _logger if _logger is not None else init_logger()
