"""Leveled logger + CHECK macros.

Capability parity with the reference Log/Logger
(include/multiverso/util/log.h:9-142, src/util/log.cpp): four levels,
``[LEVEL] [TIME] [rank]`` prefix, optional file sink, Fatal aborts, and
CHECK/CHECK_NOTNULL helpers that route through Fatal.
"""

from __future__ import annotations

import os
import sys
import time
from enum import IntEnum
from typing import Optional, TextIO


class LogLevel(IntEnum):
    DEBUG = 0
    INFO = 1
    ERROR = 2
    FATAL = 3


_LEVEL_NAMES = {LogLevel.DEBUG: "DEBUG", LogLevel.INFO: "INFO",
                LogLevel.ERROR: "ERROR", LogLevel.FATAL: "FATAL"}


class Logger:
    def __init__(self, level: LogLevel = LogLevel.INFO) -> None:
        self.level = level
        self._file: Optional[TextIO] = None
        self.kill_fatal = True

    def reset_log_level(self, level: LogLevel) -> None:
        self.level = level

    def reset_log_file(self, path: Optional[str]) -> None:
        if self._file is not None:
            self._file.close()
            self._file = None
        if path:
            self._file = open(path, "a")

    def reset_kill_fatal(self, kill: bool) -> None:
        self.kill_fatal = kill

    def _emit(self, level: LogLevel, msg: str) -> None:
        if level < self.level:
            return
        rank = os.environ.get("RANK", "0")
        ts = time.strftime("%m-%d %H:%M:%S")
        line = f"[{_LEVEL_NAMES[level]}] [{ts}] [rank {rank}] {msg}"
        from .configure import get_flag
        try:
            to_stderr = get_flag("logtostderr")
        except KeyError:
            to_stderr = False
        print(line, file=sys.stderr if to_stderr else sys.stdout, flush=True)
        if self._file is not None:
            self._file.write(line + "\n")
            self._file.flush()

    def debug(self, msg: str) -> None:
        self._emit(LogLevel.DEBUG, msg)

    def info(self, msg: str) -> None:
        self._emit(LogLevel.INFO, msg)

    def error(self, msg: str) -> None:
        self._emit(LogLevel.ERROR, msg)

    def fatal(self, msg: str) -> None:
        self._emit(LogLevel.FATAL, msg)
        if self.kill_fatal:
            # Fail-fast parity with the reference (CHECK aborts the MPI
            # job, log.h:10-13): tear down the process group so peer
            # ranks blocked in a collective fail promptly instead of
            # hanging out their 300 s timeout. destroy is best-effort —
            # a rank inside a collective cannot always be interrupted,
            # but dead TCP connections (gloo) / closed communicators
            # surface the failure to peers.
            try:
                import torch.distributed as dist
                if dist.is_initialized():
                    dist.destroy_process_group()
            except Exception:
                pass
            raise FatalError(msg)


class FatalError(RuntimeError):
    pass


log = Logger()


def CHECK(cond: bool, msg: str = "CHECK failed") -> None:
    if not cond:
        log.fatal(msg)


def CHECK_NOTNULL(x, msg: str = "CHECK_NOTNULL failed"):
    if x is None:
        log.fatal(msg)
    return x
