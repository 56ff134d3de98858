"""Moving-average throughput metric (reference utils/utils.py:52-77 parity)."""

from __future__ import annotations

from collections import deque


class Throughput:
    """Sequences/sec over a moving window of steps; tracks peak."""

    def __init__(self, window: int = 10):
        self.window = deque(maxlen=window)
        self.peak = 0.0

    def update(self, seqs: float, seconds: float):
        if seconds > 0:
            self.window.append(seqs / seconds)
            self.peak = max(self.peak, self.window[-1])

    @property
    def value(self) -> float:
        return sum(self.window) / len(self.window) if self.window else 0.0
