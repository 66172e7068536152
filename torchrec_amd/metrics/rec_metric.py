"""Metrics framework core.

Reference parity: torchrec/metrics/rec_metric.py (RecMetric :393,
RecMetricComputation :202 — torchmetrics-style local state + windowed buffers
+ all-reduce on compute) and torchrec/metrics/metric_module.py:197
(RecMetricModule).
"""

from __future__ import annotations

import abc
from enum import Enum, unique
import math
from collections import deque
from dataclasses import dataclass, field
from typing import Any, Deque, Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn


@dataclass
class RecTaskInfo:
    name: str = "DefaultTask"
    label_name: str = "label"
    prediction_name: str = "prediction"
    weight_name: str = "weight"


class WindowBuffer:
    """Sliding window of per-batch states (reference rec_metric.py:162)."""

    def __init__(self, max_size: int) -> None:
        self._max_size = max_size
        self._buffers: Deque[List[torch.Tensor]] = deque()

    def append(self, states: List[torch.Tensor]) -> None:
        self._buffers.append([s.detach().clone() for s in states])
        while len(self._buffers) > self._max_size:
            self._buffers.popleft()

    def aggregate(self, n_states: int) -> List[torch.Tensor]:
        if not self._buffers:
            return []
        out = []
        for i in range(n_states):
            out.append(torch.stack([b[i] for b in self._buffers]).sum(dim=0))
        return out


def _bsum(t: torch.Tensor) -> torch.Tensor:
    """Sum over the sample dim: [N] -> [1]; [T, N] -> [T] (fused tasks)."""
    if t.dim() > 1:
        return t.sum(dim=-1)
    return t.sum().reshape(1)


class RecMetricComputation(abc.ABC, nn.Module):
    """Accumulates local state; all-reduces at compute().

    ``n_tasks`` > 1 vectorizes the state over a leading task dim — the
    FUSED_TASKS_COMPUTATION mode (reference metrics_config.py:88
    RecComputeMode): one computation updates all tasks from stacked
    [T, N] inputs."""

    STATE_NAMES: List[str] = []

    def __init__(self, window_size: int = 100, process_group=None,
                 n_tasks: int = 1) -> None:
        super().__init__()
        self._pg = process_group
        self._n_tasks = n_tasks
        self._window = WindowBuffer(window_size)
        for name in self.STATE_NAMES:
            self.register_buffer(
                name, torch.zeros(n_tasks, dtype=torch.float64), persistent=False
            )

    def _states(self) -> List[torch.Tensor]:
        return [getattr(self, n) for n in self.STATE_NAMES]

    @abc.abstractmethod
    def update(
        self, predictions: torch.Tensor, labels: torch.Tensor, weights: Optional[torch.Tensor]
    ) -> None:
        ...

    @abc.abstractmethod
    def _compute_from(self, states: List[torch.Tensor]) -> torch.Tensor:
        ...

    def _record_window(self, batch_states: List[torch.Tensor]) -> None:
        self._window.append(batch_states)

    def _reduced(self, states: List[torch.Tensor]) -> List[torch.Tensor]:
        if self._pg is None and not (dist.is_available() and dist.is_initialized()):
            return states
        if dist.get_world_size(self._pg) <= 1:
            return states  # single rank: nothing to reduce (and the default
            # pg's backend may not cover CPU state tensors)
        pg = self._pg
        out = []
        for s in states:
            t = s.clone()
            dist.all_reduce(t, group=pg)
            out.append(t)
        return out

    def compute(self) -> Dict[str, torch.Tensor]:
        lifetime = self._reduced(self._states())
        window = self._reduced(self._window.aggregate(len(self.STATE_NAMES)) or self._states())
        return {
            "lifetime": self._compute_from(lifetime),
            "window": self._compute_from(window),
        }


def _weights_or_ones(labels: torch.Tensor, weights: Optional[torch.Tensor]) -> torch.Tensor:
    if weights is None:
        return torch.ones_like(labels, dtype=torch.float64)
    return weights.to(torch.float64)


class NEComputation(RecMetricComputation):
    """Normalized (cross-)entropy (reference metrics/ne.py)."""

    STATE_NAMES = ["cross_entropy_sum", "weighted_num_samples", "pos_labels", "neg_labels"]

    def update(self, predictions, labels, weights=None) -> None:
        p = predictions.double().clamp(1e-7, 1 - 1e-7)
        y = labels.double()
        w = _weights_or_ones(y, weights)
        ce = -(y * p.log() + (1 - y) * (1 - p).log())
        batch = [
            _bsum(ce * w),
            _bsum(w),
            _bsum(y * w),
            _bsum((1 - y) * w),
        ]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        ce_sum, n, pos, neg = states
        total = pos + neg
        base_p = (pos / total.clamp(min=1e-12)).clamp(1e-7, 1 - 1e-7)
        base_ce = -(base_p * base_p.log() + (1 - base_p) * (1 - base_p).log())
        return (ce_sum / n.clamp(min=1e-12)) / base_ce.clamp(min=1e-12)


class AUCComputation(RecMetricComputation):
    """Streaming AUC over a bounded window of raw scores (reference
    metrics/auc.py keeps full windowed tensors; we cap the buffer)."""

    STATE_NAMES: List[str] = []

    def __init__(self, window_size: int = 100, process_group=None, max_elems: int = 1_000_000):
        super().__init__(window_size, process_group)
        self._preds: List[torch.Tensor] = []
        self._labels: List[torch.Tensor] = []
        self._weights: List[torch.Tensor] = []
        self._max_elems = max_elems

    def update(self, predictions, labels, weights=None) -> None:
        self._preds.append(predictions.detach().double().reshape(-1))
        self._labels.append(labels.detach().double().reshape(-1))
        self._weights.append(_weights_or_ones(labels.reshape(-1), weights))
        total = sum(p.numel() for p in self._preds)
        dropped = False
        while total > self._max_elems and len(self._preds) > 1:
            total -= self._preds[0].numel()
            self._preds.pop(0)
            self._labels.pop(0)
            self._weights.pop(0)
            dropped = True
        if dropped and not getattr(self, "_warned_cap", False):
            import warnings

            warnings.warn(
                f"AUC score buffer exceeded max_elems={self._max_elems}; oldest "
                "batches dropped — AUC is now computed over a sliding window. "
                "Raise AUCMetric(..., max_elems=N) for a longer horizon.",
                stacklevel=2,
            )
            self._warned_cap = True

    def compute(self) -> Dict[str, torch.Tensor]:
        if not self._preds:
            return {"lifetime": torch.tensor(0.5), "window": torch.tensor(0.5)}
        p = torch.cat(self._preds)
        y = torch.cat(self._labels)
        w = torch.cat(self._weights)
        if (
            dist.is_available()
            and dist.is_initialized()
            and dist.get_world_size(self._pg) > 1
        ):
            gp = [None] * dist.get_world_size(self._pg)
            dist.all_gather_object(gp, (p, y, w), group=self._pg)
            p = torch.cat([t[0] for t in gp])
            y = torch.cat([t[1] for t in gp])
            w = torch.cat([t[2] for t in gp])
        auc = _weighted_auc(p, y, w)
        return {"lifetime": auc, "window": auc}

    def _compute_from(self, states):  # pragma: no cover - unused
        raise NotImplementedError


def _weighted_auc(preds: torch.Tensor, labels: torch.Tensor, weights: torch.Tensor) -> torch.Tensor:
    order = torch.argsort(preds, descending=True)
    y = labels[order]
    w = weights[order]
    tp = torch.cumsum(y * w, 0)
    fp = torch.cumsum((1 - y) * w, 0)
    total_pos = tp[-1] if tp.numel() else torch.tensor(0.0)
    total_neg = fp[-1] if fp.numel() else torch.tensor(0.0)
    if float(total_pos) == 0 or float(total_neg) == 0:
        return torch.tensor(0.5, dtype=torch.float64)
    tpr = torch.cat([torch.zeros(1, dtype=tp.dtype), tp / total_pos])
    fpr = torch.cat([torch.zeros(1, dtype=fp.dtype), fp / total_neg])
    return torch.trapz(tpr, fpr)


class CalibrationComputation(RecMetricComputation):
    """sum(pred)/sum(label) (reference metrics/calibration.py)."""

    STATE_NAMES = ["calibration_num", "calibration_denom"]

    def update(self, predictions, labels, weights=None) -> None:
        w = _weights_or_ones(labels.double(), weights)
        batch = [
            _bsum(predictions.double() * w),
            _bsum(labels.double() * w),
        ]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        num, denom = states
        return num / denom.clamp(min=1e-12)


class MSEComputation(RecMetricComputation):
    STATE_NAMES = ["error_sum", "weighted_num_samples"]

    def update(self, predictions, labels, weights=None) -> None:
        w = _weights_or_ones(labels.double(), weights)
        batch = [
            _bsum(w * (predictions.double() - labels.double()) ** 2),
            _bsum(w),
        ]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        err, n = states
        return err / n.clamp(min=1e-12)


@unique
class RecComputeMode(Enum):
    """Reference metrics_config.py:88."""

    UNFUSED_TASKS_COMPUTATION = "unfused"
    FUSED_TASKS_COMPUTATION = "fused"


class RecMetric(nn.Module):
    """Multi-task wrapper over a computation class (reference rec_metric.py:393).

    FUSED_TASKS_COMPUTATION runs ONE vectorized computation over stacked
    [n_tasks, N] inputs instead of one computation per task."""

    COMPUTATION: type = NEComputation
    NAME = "metric"
    FUSABLE = True

    def __init__(
        self,
        tasks: List[RecTaskInfo],
        window_size: int = 100,
        process_group=None,
        compute_mode: RecComputeMode = RecComputeMode.UNFUSED_TASKS_COMPUTATION,
        **kwargs,
    ) -> None:
        super().__init__()
        self._tasks = tasks
        self._compute_mode = compute_mode
        if compute_mode == RecComputeMode.FUSED_TASKS_COMPUTATION:
            if not self.FUSABLE:
                raise ValueError(f"{self.NAME} does not support fused-task compute")
            self._computations = nn.ModuleList(
                [
                    self.COMPUTATION(
                        window_size=window_size,
                        process_group=process_group,
                        n_tasks=len(tasks),
                        **kwargs,
                    )
                ]
            )
        else:
            self._computations = nn.ModuleList(
                [
                    self.COMPUTATION(
                        window_size=window_size, process_group=process_group, **kwargs
                    )
                    for _ in tasks
                ]
            )

    def update(
        self,
        *,
        predictions: Dict[str, torch.Tensor],
        labels: Dict[str, torch.Tensor],
        weights: Optional[Dict[str, torch.Tensor]] = None,
    ) -> None:
        if self._compute_mode == RecComputeMode.FUSED_TASKS_COMPUTATION:
            p = torch.stack([predictions[t.name] for t in self._tasks])
            y = torch.stack([labels[t.name] for t in self._tasks])
            w = (
                torch.stack(
                    [
                        weights.get(t.name, torch.ones_like(labels[t.name]))
                        for t in self._tasks
                    ]
                )
                if weights
                else None
            )
            self._computations[0].update(p, y, w)
            return
        for task, comp in zip(self._tasks, self._computations):
            comp.update(
                predictions[task.name],
                labels[task.name],
                weights.get(task.name) if weights else None,
            )

    def compute(self) -> Dict[str, torch.Tensor]:
        out = {}
        if self._compute_mode == RecComputeMode.FUSED_TASKS_COMPUTATION:
            res = self._computations[0].compute()
            for i, task in enumerate(self._tasks):
                out[f"{self.NAME}-{task.name}|lifetime_{self.NAME}"] = res["lifetime"][i]
                out[f"{self.NAME}-{task.name}|window_{self.NAME}"] = res["window"][i]
            return out
        for task, comp in zip(self._tasks, self._computations):
            res = comp.compute()
            out[f"{self.NAME}-{task.name}|lifetime_{self.NAME}"] = res["lifetime"]
            out[f"{self.NAME}-{task.name}|window_{self.NAME}"] = res["window"]
        return out


class NEMetric(RecMetric):
    COMPUTATION = NEComputation
    NAME = "ne"


class AUCMetric(RecMetric):
    COMPUTATION = AUCComputation
    NAME = "auc"
    FUSABLE = False  # keeps raw score buffers, not sum-states


class CalibrationMetric(RecMetric):
    COMPUTATION = CalibrationComputation
    NAME = "calibration"


class MSEMetric(RecMetric):
    COMPUTATION = MSEComputation
    NAME = "mse"


class AccuracyComputation(RecMetricComputation):
    """Thresholded accuracy (reference metrics/accuracy.py)."""

    STATE_NAMES = ["accuracy_sum", "weighted_num_samples"]

    def update(self, predictions, labels, weights=None) -> None:
        y = labels.double()
        w = _weights_or_ones(y, weights)
        hit = ((predictions.double() >= 0.5) == (y >= 0.5)).double()
        batch = [_bsum(hit * w), _bsum(w)]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        acc, n = states
        return acc / n.clamp(min=1e-12)


class PrecisionComputation(RecMetricComputation):
    """Precision at threshold 0.5 (reference metrics/precision.py)."""

    STATE_NAMES = ["true_pos_sum", "false_pos_sum"]

    def update(self, predictions, labels, weights=None) -> None:
        y = labels.double()
        w = _weights_or_ones(y, weights)
        pred_pos = (predictions.double() >= 0.5).double()
        batch = [
            _bsum(pred_pos * y * w),
            _bsum(pred_pos * (1 - y) * w),
        ]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        tp, fp = states
        return tp / (tp + fp).clamp(min=1e-12)


class RecallComputation(RecMetricComputation):
    """Recall at threshold 0.5 (reference metrics/recall.py)."""

    STATE_NAMES = ["true_pos_sum", "false_neg_sum"]

    def update(self, predictions, labels, weights=None) -> None:
        y = labels.double()
        w = _weights_or_ones(y, weights)
        pred_pos = (predictions.double() >= 0.5).double()
        batch = [
            _bsum(pred_pos * y * w),
            _bsum((1 - pred_pos) * y * w),
        ]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        tp, fn = states
        return tp / (tp + fn).clamp(min=1e-12)


def grouped_auc(
    predictions: torch.Tensor,
    labels: torch.Tensor,
    group_ids: torch.Tensor,
    weights: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """GAUC: mean per-group AUC over groups with both classes (reference
    metrics/gauc.py compute_gauc_3d — jagged per-session AUC)."""
    out = []
    for g in torch.unique(group_ids):
        m = group_ids == g
        y = labels[m]
        if y.numel() < 2 or y.min() == y.max():
            continue
        w = _weights_or_ones(y.double(), weights[m] if weights is not None else None)
        out.append(_weighted_auc(predictions[m].double(), y.double(), w))
    if not out:
        return torch.tensor(0.5, dtype=torch.float64)
    return torch.stack(out).mean()


class AccuracyMetric(RecMetric):
    COMPUTATION = AccuracyComputation
    NAME = "accuracy"


class PrecisionMetric(RecMetric):
    COMPUTATION = PrecisionComputation
    NAME = "precision"


class RecallMetric(RecMetric):
    COMPUTATION = RecallComputation
    NAME = "recall"
