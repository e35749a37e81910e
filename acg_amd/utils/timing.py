"""Wall-clock timing helpers (reference: acg/time.h gettime()/elapsed())."""

from __future__ import annotations

import time


def gettime() -> float:
    """Monotonic wall-clock seconds (reference acg/time.h:57-93)."""
    return time.perf_counter()


def elapsed(t0: float, t1: float | None = None) -> float:
    """Seconds elapsed since ``t0`` (until ``t1`` if given)."""
    return (time.perf_counter() if t1 is None else t1) - t0


class Timer:
    """Accumulating timer for per-op statistics."""

    __slots__ = ("seconds", "count", "_t0")

    def __init__(self):
        self.seconds = 0.0
        self.count = 0
        self._t0 = 0.0

    def start(self):
        self._t0 = time.perf_counter()

    def stop(self):
        self.seconds += time.perf_counter() - self._t0
        self.count += 1

    def __enter__(self):
        self.start()
        return self

    def __exit__(self, *a):
        self.stop()
        return False
