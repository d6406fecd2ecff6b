"""Fixed-window moving average (reference: internal/movingaverage/simple.go).

Decays to zero as zeros are fed in — this is what enables scale-to-zero.
"""
from __future__ import annotations


class SimpleMovingAverage:
    def __init__(self, window_count: int, seed: float = 0.0):
        assert window_count > 0
        self._values = [seed] * window_count
        self._idx = 0
        self._filled = seed != 0.0

    def next(self, value: float) -> float:
        self._values[self._idx] = value
        self._idx = (self._idx + 1) % len(self._values)
        return self.calculate()

    def calculate(self) -> float:
        return sum(self._values) / len(self._values)

    def history(self) -> list[float]:
        return list(self._values)

    def load(self, values: list[float]) -> None:
        for i, v in enumerate(values[: len(self._values)]):
            self._values[i] = v
