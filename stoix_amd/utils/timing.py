"""Rolling-window wall-clock timers (parity with
/root/reference/stoix/utils/timing_utils.py:8-133, `TimingTracker`)."""
from __future__ import annotations

import time
from collections import defaultdict, deque
from contextlib import contextmanager
from typing import Dict


class TimingTracker:
    def __init__(self, maxlen: int = 10):
        self.maxlen = maxlen
        self._times: Dict[str, deque] = defaultdict(lambda: deque(maxlen=self.maxlen))

    @contextmanager
    def time(self, name: str):
        t0 = time.perf_counter()
        try:
            yield
        finally:
            self._times[name].append(time.perf_counter() - t0)

    def add(self, name: str, seconds: float) -> None:
        self._times[name].append(seconds)

    def mean(self, name: str) -> float:
        d = self._times[name]
        return sum(d) / len(d) if d else 0.0

    def latest(self, name: str) -> float:
        d = self._times[name]
        return d[-1] if d else 0.0

    def summary(self, prefix: str = "time/") -> Dict[str, float]:
        return {f"{prefix}{k}": self.mean(k) for k in self._times}
