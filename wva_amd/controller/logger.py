"""Structured JSON logger (zap-analog).

Parity with /root/reference/internal/logger/logger.go: singleton, JSON
lines to stdout, level from the LOG_LEVEL env var.
"""

from __future__ import annotations

import json
import logging
import os
import sys
import time
from typing import Optional


class _JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        entry = {
            "level": record.levelname.lower(),
            "ts": time.time(),
            "msg": record.getMessage(),
        }
        extra = getattr(record, "kv", None)
        if extra:
            entry.update(extra)
        if record.exc_info and record.exc_info[0] is not None:
            entry["error"] = str(record.exc_info[1])
        return json.dumps(entry)


def level_from_env() -> int:
    return {
        "debug": logging.DEBUG,
        "info": logging.INFO,
        "warn": logging.WARNING,
        "error": logging.ERROR,
    }.get(os.environ.get("LOG_LEVEL", "").lower(), logging.INFO)


class _Logger:
    def __init__(self) -> None:
        self._logger: Optional[logging.Logger] = None

    def init(self, level: Optional[int] = None) -> None:
        logger = logging.getLogger("wva_amd")
        logger.handlers.clear()
        handler = logging.StreamHandler(sys.stdout)
        handler.setFormatter(_JsonFormatter())
        logger.addHandler(handler)
        logger.setLevel(level if level is not None else level_from_env())
        logger.propagate = False
        self._logger = logger

    def _get(self) -> logging.Logger:
        if self._logger is None:
            self.init()
        return self._logger

    def _log(self, level: int, msg: str, kv: dict) -> None:
        self._get().log(level, msg, extra={"kv": kv})

    def debug(self, msg: str, **kv) -> None:
        self._log(logging.DEBUG, msg, kv)

    def info(self, msg: str, **kv) -> None:
        self._log(logging.INFO, msg, kv)

    def warn(self, msg: str, **kv) -> None:
        self._log(logging.WARNING, msg, kv)

    def error(self, msg: str, **kv) -> None:
        self._log(logging.ERROR, msg, kv)


log = _Logger()
