"""Benchmark statistics (reference: bin/statistics.{hpp,cpp} — notably the
trimean (q1 + 2*q2 + q3)/4 used for all reported timings)."""
from __future__ import annotations

import math
from typing import List


class Statistics:
    def __init__(self, samples: List[float] = None):
        self.samples: List[float] = list(samples) if samples else []

    def insert(self, v: float):
        self.samples.append(v)

    def count(self) -> int:
        return len(self.samples)

    def min(self) -> float:
        return min(self.samples)

    def max(self) -> float:
        return max(self.samples)

    def avg(self) -> float:
        return sum(self.samples) / len(self.samples)

    def stddev(self) -> float:
        m = self.avg()
        return math.sqrt(sum((s - m) ** 2 for s in self.samples) / len(self.samples))

    def _quantile(self, q: float) -> float:
        s = sorted(self.samples)
        if len(s) == 1:
            return s[0]
        pos = q * (len(s) - 1)
        lo = int(math.floor(pos))
        hi = min(lo + 1, len(s) - 1)
        frac = pos - lo
        return s[lo] * (1 - frac) + s[hi] * frac

    def med(self) -> float:
        return self._quantile(0.5)

    def trimean(self) -> float:
        return (self._quantile(0.25) + 2 * self._quantile(0.5) + self._quantile(0.75)) / 4
