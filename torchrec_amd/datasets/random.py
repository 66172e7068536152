"""Synthetic random recsys dataset + Batch type.

Reference parity: torchrec/datasets/random.py:125 (RandomRecDataset) and
torchrec/datasets/utils.py (Batch).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Iterator, List, Optional

import torch
from torch.utils.data import IterableDataset

from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor
from torchrec_amd.streamable import Pipelineable


@dataclass
class Batch(Pipelineable):
    dense_features: torch.Tensor
    sparse_features: KeyedJaggedTensor
    labels: torch.Tensor

    def to(self, device: torch.device, non_blocking: bool = False) -> "Batch":
        return Batch(
            dense_features=self.dense_features.to(device=device, non_blocking=non_blocking),
            sparse_features=self.sparse_features.to(device=device, non_blocking=non_blocking),
            labels=self.labels.to(device=device, non_blocking=non_blocking),
        )

    def record_stream(self, stream: torch.Stream) -> None:
        self.dense_features.record_stream(stream)
        self.sparse_features.record_stream(stream)
        self.labels.record_stream(stream)

    def pin_memory(self) -> "Batch":
        return Batch(
            dense_features=self.dense_features.pin_memory(),
            sparse_features=self.sparse_features.pin_memory(),
            labels=self.labels.pin_memory(),
        )


def generate_batch(
    keys: List[str],
    batch_size: int,
    hash_sizes: List[int],
    ids_per_feature: int = 20,
    num_dense: int = 13,
    pooling_avg: Optional[int] = None,
    device: Optional[torch.device] = None,
    generator: Optional[torch.Generator] = None,
    pinned: bool = False,
    learnable_labels: bool = False,
) -> Batch:
    """One synthetic Criteo-shaped batch (random ids / dense / labels).

    ``learnable_labels`` makes the label a deterministic function of the ids
    (parity of the first feature's id) so a training loop has signal to fit —
    used by the convergence tests."""
    device = device or torch.device("cpu")
    lengths_list = []
    values_list = []
    for k, hs in zip(keys, hash_sizes):
        if pooling_avg is None:
            lengths = torch.full((batch_size,), ids_per_feature, dtype=torch.int64, device=device)
        else:
            lengths = torch.poisson(
                torch.full((batch_size,), float(pooling_avg), device=device)
            ).to(torch.int64)
        n = int(lengths.sum())
        values = torch.randint(0, hs, (n,), dtype=torch.int64, device=device, generator=generator)
        lengths_list.append(lengths)
        values_list.append(values)
    kjt = KeyedJaggedTensor(
        keys=keys,
        values=torch.cat(values_list) if values_list else torch.empty(0, dtype=torch.int64),
        lengths=torch.cat(lengths_list) if lengths_list else torch.empty(0, dtype=torch.int64),
        stride=batch_size,
    )
    if learnable_labels and values_list and values_list[0].numel() >= batch_size:
        # label = parity of the sample's first id of feature 0 (lengths must
        # be >=1 for f0; callers pass fixed lengths for convergence tests)
        first = values_list[0][: batch_size]
        labels = (first % 2).to(torch.int64)
    else:
        labels = torch.randint(0, 2, (batch_size,), device=device, generator=generator)
    batch = Batch(
        dense_features=torch.rand(batch_size, num_dense, device=device, generator=generator),
        sparse_features=kjt,
        labels=labels,
    )
    if pinned and device.type == "cpu":
        batch = batch.pin_memory()
    return batch


class RandomRecDataset(IterableDataset):
    """Infinite synthetic dataset (reference: torchrec/datasets/random.py:125)."""

    def __init__(
        self,
        keys: List[str],
        batch_size: int,
        hash_sizes: List[int],
        ids_per_feature: int = 20,
        num_dense: int = 13,
        pooling_avg: Optional[int] = None,
        num_batches: Optional[int] = None,
        seed: int = 0,
        learnable_labels: bool = False,
    ) -> None:
        self.learnable_labels = learnable_labels
        super().__init__()
        self.keys = keys
        self.batch_size = batch_size
        self.hash_sizes = hash_sizes
        self.ids_per_feature = ids_per_feature
        self.num_dense = num_dense
        self.pooling_avg = pooling_avg
        self.num_batches = num_batches
        self.seed = seed

    def __iter__(self) -> Iterator[Batch]:
        gen = torch.Generator().manual_seed(self.seed)
        i = 0
        while self.num_batches is None or i < self.num_batches:
            yield generate_batch(
                self.keys,
                self.batch_size,
                self.hash_sizes,
                ids_per_feature=self.ids_per_feature,
                num_dense=self.num_dense,
                pooling_avg=self.pooling_avg,
                generator=gen,
                learnable_labels=self.learnable_labels,
            )
            i += 1
