"""RecMetricModule + ThroughputMetric.

Reference parity: torchrec/metrics/metric_module.py:197 and
torchrec/metrics/throughput.py:35.
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

import torch
import torch.nn as nn

from torchrec_amd.metrics.rec_metric import RecMetric, RecTaskInfo


class ThroughputMetric(nn.Module):
    """Examples/sec from wall clock between updates (reference throughput.py:35)."""

    def __init__(self, batch_size: int, world_size: int, window_seconds: int = 100) -> None:
        super().__init__()
        self._batch_size = batch_size
        self._world_size = world_size
        self._window_seconds = window_seconds
        self._steps = 0
        self._t0: Optional[float] = None
        self._ticks: List[float] = []

    def update(self) -> None:
        now = time.monotonic()
        if self._t0 is None:
            self._t0 = now
        self._steps += 1
        self._ticks.append(now)
        cutoff = now - self._window_seconds
        while len(self._ticks) > 2 and self._ticks[0] < cutoff:
            self._ticks.pop(0)

    def compute(self) -> Dict[str, torch.Tensor]:
        if self._t0 is None or self._steps < 2:
            return {"throughput-throughput|total_examples": torch.tensor(0.0)}
        lifetime = (
            self._steps * self._batch_size * self._world_size / (time.monotonic() - self._t0)
        )
        window_span = self._ticks[-1] - self._ticks[0] if len(self._ticks) > 1 else 1.0
        window = (len(self._ticks) - 1) * self._batch_size * self._world_size / max(
            window_span, 1e-9
        )
        return {
            "throughput-throughput|lifetime_throughput": torch.tensor(lifetime),
            "throughput-throughput|window_throughput": torch.tensor(window),
            "throughput-throughput|total_examples": torch.tensor(
                float(self._steps * self._batch_size * self._world_size)
            ),
        }


class RecMetricModule(nn.Module):
    """Batches RecMetrics + throughput (reference metric_module.py:197)."""

    def __init__(
        self,
        batch_size: int,
        world_size: int,
        rec_tasks: Optional[List[RecTaskInfo]] = None,
        rec_metrics: Optional[List[RecMetric]] = None,
        throughput_metric: Optional[ThroughputMetric] = None,
        compute_interval_steps: int = 100,
    ) -> None:
        super().__init__()
        self.rec_tasks = rec_tasks or []
        self.rec_metrics = nn.ModuleList(rec_metrics or [])
        self.throughput_metric = throughput_metric
        self.compute_interval_steps = compute_interval_steps
        self.trained_batches = 0

    def update(
        self,
        *,
        predictions: Dict[str, torch.Tensor],
        labels: Dict[str, torch.Tensor],
        weights: Optional[Dict[str, torch.Tensor]] = None,
    ) -> None:
        for metric in self.rec_metrics:
            metric.update(predictions=predictions, labels=labels, weights=weights)
        if self.throughput_metric is not None:
            self.throughput_metric.update()
        self.trained_batches += 1

    def should_compute(self) -> bool:
        return self.trained_batches % self.compute_interval_steps == 0

    def compute(self) -> Dict[str, torch.Tensor]:
        out: Dict[str, torch.Tensor] = {}
        for metric in self.rec_metrics:
            out.update(metric.compute())
        if self.throughput_metric is not None:
            out.update(self.throughput_metric.compute())
        return out
