"""Latency percentile tracker (reference: torchrec/utils/percentile_logger.py:17)."""

from __future__ import annotations

from typing import Dict, List


class PercentileLogger:
    def __init__(self, name: str = "", max_samples: int = 100_000) -> None:
        self._name = name
        self._samples: List[float] = []
        self._max = max_samples

    def add(self, value: float) -> None:
        if len(self._samples) >= self._max:
            self._samples.pop(0)
        self._samples.append(value)

    def percentile(self, p: float) -> float:
        if not self._samples:
            return 0.0
        s = sorted(self._samples)
        idx = min(len(s) - 1, int(round(p / 100.0 * (len(s) - 1))))
        return s[idx]

    def summary(self) -> Dict[str, float]:
        return {
            "p50": self.percentile(50),
            "p90": self.percentile(90),
            "p99": self.percentile(99),
            "count": float(len(self._samples)),
        }

    def reset(self) -> None:
        self._samples.clear()
