"""Bunyan-compatible structured JSON logging.

The reference logs through node-bunyan everywhere, with child loggers carrying
bound fields (e.g. ``lib/zookeeperMgr.js:70``, ``lib/postgresMgr.js:262``) and
``-v`` flags stacking the level down to TRACE (``sitter.js:61-66``).  This
module reproduces that: one-line JSON records with bunyan's field set
(``v/level/name/hostname/pid/time/msg``), numeric bunyan levels, and cheap
child loggers.  Output is line-buffered to a stream or file so logs survive a
``kill -9`` of the daemon (the failure mode this whole system exists for).
"""

from __future__ import annotations

import json
import os
import socket
import sys
import threading
import time
from typing import Any, IO, Optional

TRACE = 10
DEBUG = 20
INFO = 30
WARN = 40
ERROR = 50
FATAL = 60

_LEVEL_NAMES = {
    "trace": TRACE, "debug": DEBUG, "info": INFO,
    "warn": WARN, "error": ERROR, "fatal": FATAL,
}
_NAMES_BY_LEVEL = {v: k for k, v in _LEVEL_NAMES.items()}

_HOSTNAME = socket.gethostname()


def resolve_level(level: "int | str") -> int:
    if isinstance(level, int):
        return level
    try:
        return _LEVEL_NAMES[level.lower()]
    except KeyError:
        raise ValueError("unknown log level %r" % (level,))


def level_from_verbosity(nverbose: int) -> int:
    """``-v`` stacking: 0 → INFO, 1 → DEBUG, >=2 → TRACE (ref sitter.js:61-66)."""
    if nverbose <= 0:
        return INFO
    if nverbose == 1:
        return DEBUG
    return TRACE


class _Sink:
    """A shared, thread-safe output stream for a logger tree."""

    def __init__(self, stream: Optional[IO[str]] = None, path: Optional[str] = None):
        self._lock = threading.Lock()
        self._path = path
        if path is not None:
            self._stream = open(path, "a", buffering=1)
        else:
            self._stream = stream if stream is not None else sys.stderr

    def write_line(self, line: str) -> None:
        with self._lock:
            try:
                self._stream.write(line + "\n")
                self._stream.flush()
            except (ValueError, OSError):
                pass  # stream closed during shutdown; logging must never throw


class Logger:
    """A bunyan-style logger.  ``child(component=...)`` binds fields."""

    def __init__(self, name: str, level: "int | str" = INFO,
                 stream: Optional[IO[str]] = None, path: Optional[str] = None,
                 _sink: Optional[_Sink] = None, _fields: Optional[dict] = None):
        self.name = name
        self.level = resolve_level(level)
        self._sink = _sink if _sink is not None else _Sink(stream, path)
        self._fields = dict(_fields or {})

    # -- structure ----------------------------------------------------------
    def child(self, **fields: Any) -> "Logger":
        merged = dict(self._fields)
        merged.update(fields)
        return Logger(self.name, self.level, _sink=self._sink, _fields=merged)

    def set_level(self, level: "int | str") -> None:
        self.level = resolve_level(level)

    def is_enabled(self, level: int) -> bool:
        return level >= self.level

    # -- emission -----------------------------------------------------------
    def _emit(self, level: int, msg: str, extra: dict) -> None:
        if level < self.level:
            return
        rec: dict = {
            "name": self.name,
            "hostname": _HOSTNAME,
            "pid": os.getpid(),
            "level": level,
        }
        rec.update(self._fields)
        for k, v in extra.items():
            if isinstance(v, BaseException):
                rec[k] = {"message": str(v), "name": type(v).__name__}
            else:
                rec[k] = v
        rec["msg"] = msg
        rec["time"] = time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime()) + \
            (".%03dZ" % (int(time.time() * 1000) % 1000))
        rec["v"] = 0
        try:
            line = json.dumps(rec, default=repr, separators=(",", ":"))
        except Exception:
            line = json.dumps({"name": self.name, "level": level, "msg": msg,
                               "v": 0}, separators=(",", ":"))
        self._sink.write_line(line)

    def trace(self, msg: str, **extra: Any) -> None:
        self._emit(TRACE, msg, extra)

    def debug(self, msg: str, **extra: Any) -> None:
        self._emit(DEBUG, msg, extra)

    def info(self, msg: str, **extra: Any) -> None:
        self._emit(INFO, msg, extra)

    def warn(self, msg: str, **extra: Any) -> None:
        self._emit(WARN, msg, extra)

    def error(self, msg: str, **extra: Any) -> None:
        self._emit(ERROR, msg, extra)

    def fatal(self, msg: str, **extra: Any) -> None:
        self._emit(FATAL, msg, extra)


def null_logger(name: str = "null") -> Logger:
    lg = Logger(name, level=FATAL + 1)
    return lg
