"""Structured logging with the reference's log_tag convention (§5.5:
zerolog fields namespacing subsystems — rw_pool, rw_edge, chunk_pb,
null_validation, FOCUS...)."""
from __future__ import annotations

import json
import logging
import sys
import time

_LEVELS = {
    "trace": 5, "debug": logging.DEBUG, "info": logging.INFO,
    "warn": logging.WARNING, "error": logging.ERROR,
    "fatal": logging.CRITICAL,
}


class TaggedLogger:
    def __init__(self, name: str, level: str = "info", stream=None):
        self.name = name
        self.level = _LEVELS.get(level, logging.INFO)
        self.stream = stream or sys.stderr

    def _emit(self, lvl: str, msg: str, **fields):
        if _LEVELS.get(lvl, 0) < self.level:
            return
        rec = {"ts": round(time.time(), 3), "level": lvl,
               "logger": self.name, "msg": msg}
        rec.update(fields)
        print(json.dumps(rec, default=str), file=self.stream, flush=True)

    def debug(self, msg, **f):
        self._emit("debug", msg, **f)

    def info(self, msg, **f):
        self._emit("info", msg, **f)

    def warn(self, msg, **f):
        self._emit("warn", msg, **f)

    def error(self, msg, **f):
        self._emit("error", msg, **f)


_loggers = {}


def get_logger(name: str, level: str = "info") -> TaggedLogger:
    if name not in _loggers:
        _loggers[name] = TaggedLogger(name, level)
    return _loggers[name]
