"""Extended metric zoo (reference: torchrec/metrics/{ndcg,recall_session,
mae,ctr,weighted_avg,cross_entropy,tower_qps}.py).

Session metrics (NDCG, recall@k) take a ``session_ids`` tensor aligned with
predictions; sessions may arrive in any order within the batch.
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

import torch

from torchrec_amd.metrics.rec_metric import (
    RecMetric,
    RecMetricComputation,
    RecTaskInfo,
    _bsum,
    _weights_or_ones,
)


class MAEComputation(RecMetricComputation):
    """Weighted mean absolute error (reference metrics/mae.py)."""

    STATE_NAMES = ["error_sum", "weighted_num_samples"]

    def update(self, predictions, labels, weights=None) -> None:
        w = _weights_or_ones(labels.double(), weights)
        batch = [
            _bsum(w * (predictions.double() - labels.double()).abs()),
            _bsum(w),
        ]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        err, n = states
        return err / n.clamp(min=1e-12)


class CTRComputation(RecMetricComputation):
    """Weighted label mean — click-through rate (reference metrics/ctr.py)."""

    STATE_NAMES = ["ctr_num", "ctr_denom"]

    def update(self, predictions, labels, weights=None) -> None:
        w = _weights_or_ones(labels.double(), weights)
        batch = [_bsum(labels.double() * w), _bsum(w)]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        num, denom = states
        return num / denom.clamp(min=1e-12)


class WeightedAvgComputation(RecMetricComputation):
    """Weighted prediction mean (reference metrics/weighted_avg.py)."""

    STATE_NAMES = ["weighted_sum", "weighted_num_samples"]

    def update(self, predictions, labels, weights=None) -> None:
        w = _weights_or_ones(labels.double(), weights)
        batch = [_bsum(predictions.double() * w), _bsum(w)]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        num, denom = states
        return num / denom.clamp(min=1e-12)


class LogLossComputation(RecMetricComputation):
    """Un-normalized binary cross entropy (reference metrics/cross_entropy)."""

    STATE_NAMES = ["ce_sum", "weighted_num_samples"]

    def update(self, predictions, labels, weights=None) -> None:
        p = predictions.double().clamp(1e-7, 1 - 1e-7)
        y = labels.double()
        w = _weights_or_ones(y, weights)
        ce = -(y * p.log() + (1 - y) * (1 - p).log())
        batch = [_bsum(ce * w), _bsum(w)]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        ce, n = states
        return ce / n.clamp(min=1e-12)


class MAEMetric(RecMetric):
    COMPUTATION = MAEComputation
    NAME = "mae"


class CTRMetric(RecMetric):
    COMPUTATION = CTRComputation
    NAME = "ctr"


class WeightedAvgMetric(RecMetric):
    COMPUTATION = WeightedAvgComputation
    NAME = "weighted_avg"


class LogLossMetric(RecMetric):
    COMPUTATION = LogLossComputation
    NAME = "logloss"


# ---------------------------------------------------------------------------
# session metrics
# ---------------------------------------------------------------------------


def _session_sort(predictions: torch.Tensor, session_ids: torch.Tensor):
    """Stable (session, -prediction) ordering + per-element within-session
    rank. Returns (perm, session_index, rank, counts)."""
    order = torch.argsort(predictions, descending=True, stable=True)
    sessions_sorted, order2 = torch.sort(session_ids[order], stable=True)
    perm = order[order2]
    uniq, inverse, counts = torch.unique_consecutive(
        sessions_sorted, return_inverse=True, return_counts=True
    )
    starts = torch.cumsum(counts, 0) - counts
    rank = torch.arange(perm.numel(), device=perm.device) - starts[inverse]
    return perm, inverse, rank, counts


class NDCGComputation(RecMetricComputation):
    """Session NDCG with exponential gain (reference metrics/ndcg.py)."""

    STATE_NAMES = ["ndcg_sum", "num_sessions"]

    def update(self, predictions, labels, weights=None, session_ids=None) -> None:
        assert session_ids is not None, "NDCG needs session_ids"
        y = labels.double()
        perm, sess_idx, rank, counts = _session_sort(predictions.double(), session_ids)
        n_sessions = counts.numel()
        discount = 1.0 / torch.log2(rank.double() + 2.0)
        gains = (torch.pow(2.0, y[perm]) - 1.0) * discount
        dcg = torch.zeros(n_sessions, dtype=torch.float64)
        dcg.index_add_(0, sess_idx, gains)
        # ideal ordering: sort labels desc within session
        perm_i, sess_i, rank_i, _ = _session_sort(y, session_ids)
        gains_i = (torch.pow(2.0, y[perm_i]) - 1.0) / torch.log2(rank_i.double() + 2.0)
        idcg = torch.zeros(n_sessions, dtype=torch.float64)
        idcg.index_add_(0, sess_i, gains_i)
        valid = idcg > 0
        ndcg = torch.where(valid, dcg / idcg.clamp(min=1e-12), torch.zeros_like(dcg))
        batch = [
            ndcg[valid].sum().reshape(1),
            valid.double().sum().reshape(1),
        ]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        s, n = states
        return s / n.clamp(min=1e-12)


class RecallSessionComputation(RecMetricComputation):
    """Session recall@k: positives ranked in the session top-k over total
    positives (reference metrics/recall_session.py)."""

    STATE_NAMES = ["recall_num", "recall_denom"]

    def __init__(self, window_size: int = 100, process_group=None,
                 n_tasks: int = 1, top_k: int = 10):
        super().__init__(window_size, process_group, n_tasks)
        self._top_k = top_k

    def update(self, predictions, labels, weights=None, session_ids=None) -> None:
        assert session_ids is not None, "session recall needs session_ids"
        y = (labels.double() > 0).double()
        perm, sess_idx, rank, counts = _session_sort(predictions.double(), session_ids)
        in_topk = (rank < self._top_k).double()
        hits = torch.zeros(counts.numel(), dtype=torch.float64)
        hits.index_add_(0, sess_idx, y[perm] * in_topk)
        pos = torch.zeros(counts.numel(), dtype=torch.float64)
        pos.index_add_(0, sess_idx, y[perm])
        batch = [hits.sum().reshape(1), pos.sum().reshape(1)]
        for name, b in zip(self.STATE_NAMES, batch):
            getattr(self, name).add_(b)
        self._record_window(batch)

    def _compute_from(self, states) -> torch.Tensor:
        num, denom = states
        return num / denom.clamp(min=1e-12)


class _SessionMetric(RecMetric):
    FUSABLE = False

    def update(  # type: ignore[override]
        self,
        *,
        predictions: Dict[str, torch.Tensor],
        labels: Dict[str, torch.Tensor],
        weights: Optional[Dict[str, torch.Tensor]] = None,
        session_ids: Optional[Dict[str, torch.Tensor]] = None,
    ) -> None:
        for task, comp in zip(self._tasks, self._computations):
            comp.update(
                predictions[task.name],
                labels[task.name],
                weights.get(task.name) if weights else None,
                session_ids[task.name] if session_ids else None,
            )


class NDCGMetric(_SessionMetric):
    COMPUTATION = NDCGComputation
    NAME = "ndcg"


class RecallSessionMetric(_SessionMetric):
    COMPUTATION = RecallSessionComputation
    NAME = "recall_session"


class TowerQPSMetric(torch.nn.Module):
    """Per-tower examples/sec (reference metrics/tower_qps.py): counts
    examples per named tower against wall time between warmup and now."""

    def __init__(self, towers: List[str], warmup_steps: int = 2) -> None:
        super().__init__()
        self._towers = towers
        self._warmup_steps = warmup_steps
        self._steps = 0
        self._counts = {t: 0 for t in towers}
        self._t0: Optional[float] = None

    def update(self, batch_sizes: Dict[str, int]) -> None:
        self._steps += 1
        if self._steps <= self._warmup_steps:
            return
        if self._t0 is None:
            self._t0 = time.perf_counter()
            return  # first timed step starts the clock; count from next
        for t, n in batch_sizes.items():
            if t in self._counts:
                self._counts[t] += n

    def compute(self) -> Dict[str, float]:
        if self._t0 is None:
            return {f"tower_qps-{t}": 0.0 for t in self._towers}
        dt = max(time.perf_counter() - self._t0, 1e-9)
        return {f"tower_qps-{t}": self._counts[t] / dt for t in self._towers}
