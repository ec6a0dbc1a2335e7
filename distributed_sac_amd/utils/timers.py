"""Step timing / throughput instrumentation (SURVEY §5.1 — absent in the
reference; first-class here for the bench + rocprof workflow)."""

from __future__ import annotations

import time
from collections import deque
from typing import Deque, Dict


class StepTimer:
    """Sliding-window rate meter (events/sec over the last N marks)."""

    def __init__(self, window: int = 200):
        self.window = window
        self._marks: Deque[float] = deque(maxlen=window)
        self._counts: Deque[int] = deque(maxlen=window)
        self.total = 0

    def mark(self, n: int = 1) -> None:
        self._marks.append(time.perf_counter())
        self._counts.append(n)
        self.total += n

    def rate(self) -> float:
        if len(self._marks) < 2:
            return 0.0
        dt = self._marks[-1] - self._marks[0]
        if dt <= 0:
            return 0.0
        return sum(list(self._counts)[1:]) / dt


class Timers:
    def __init__(self):
        self.t: Dict[str, StepTimer] = {}

    def __getitem__(self, k: str) -> StepTimer:
        if k not in self.t:
            self.t[k] = StepTimer()
        return self.t[k]
