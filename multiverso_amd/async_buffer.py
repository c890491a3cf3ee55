"""ASyncBuffer — double-buffered prefetcher.

Capability parity with the reference ASyncBuffer<T>
(include/multiverso/util/async_buffer.h:11-116): ``get()`` returns the
ready buffer and immediately triggers a background fill of the other one;
the fill function runs on a worker thread (the reference's model-pull
pipelining utility; LogReg's pipeline mode hand-rolls the same pattern,
ps_model.cpp:236-271).

MI355X note: when the fill function issues table ops they are collectives
— every rank's ASyncBuffer must issue fills in the same order, which holds
for the lockstep block pipelines that use it."""

from __future__ import annotations

import threading
from typing import Callable, Generic, List, TypeVar

T = TypeVar("T")


class ASyncBuffer(Generic[T]):
    def __init__(self, buffer0: T, buffer1: T,
                 fill: Callable[[T], None]) -> None:
        self._buffers: List[T] = [buffer0, buffer1]
        self._fill = fill
        self._ready = 0
        self._thread: threading.Thread | None = None
        self._start_fill(0)

    def _start_fill(self, idx: int) -> None:
        def run() -> None:
            self._fill(self._buffers[idx])

        self._thread = threading.Thread(target=run, daemon=True)
        self._thread.start()

    def get(self) -> T:
        """Block until the in-flight fill completes, return that buffer,
        and prefetch into the other (async_buffer.h:31-40)."""
        assert self._thread is not None
        self._thread.join()
        ready = self._ready
        self._ready = 1 - ready
        self._start_fill(self._ready)
        return self._buffers[ready]

    def wait(self) -> None:
        if self._thread is not None:
            self._thread.join()
