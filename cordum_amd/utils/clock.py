"""Injectable clock so reconciler/TTL/backoff semantics are unit-testable
without sleeping (the reference tests achieve this with miniredis FastForward)."""
from __future__ import annotations

import time


class Clock:
    def now(self) -> float:  # seconds, float
        raise NotImplementedError

    def now_micros(self) -> int:
        return int(self.now() * 1_000_000)

    def now_millis(self) -> int:
        return int(self.now() * 1000)


class SystemClock(Clock):
    def now(self) -> float:
        return time.time()


class ManualClock(Clock):
    def __init__(self, start: float = 1_700_000_000.0):
        self._t = start

    def now(self) -> float:
        return self._t

    def advance(self, seconds: float) -> None:
        self._t += seconds


SYSTEM_CLOCK = SystemClock()
